"""Python surface of the hand-written CDNA4 kernel extension ``_mzk``.

This is the MI355X-native replacement for the reference's runtime-compiled
OpenCL/CUDA kernel sets (znicz/ocl/*.cl + znicz/cuda/*.cu) and the
opencl4py/cuda4py binding layer: every kernel is precompiled into one
in-tree HIP extension (hipcc --offload-arch=gfx950) and called through
thin wrappers here.  On a GPU box the extension is REQUIRED — ops fail
loudly instead of falling back to eager PyTorch.

Build: ``python veles/znicz_amd/ops/setup.py build_ext --inplace`` (or
``__graft_entry__.build()`` at the repo root).
"""

_ext = None
_ext_err = None


def _load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        import torch  # noqa: F401  (must be imported before the ext .so)
        from veles.znicz_amd.ops import _mzk  # built in-tree
        _ext = _mzk
        _apply_routing(_ext)
    except ImportError as e:
        _ext_err = e
    return _ext


def _apply_routing(mod):
    """Load the swept gfx950 routing table (SURVEY §2b autotune-DB row)
    and override the kernel launchers' built-in heuristics.  Regenerate
    with tools/gen_routing.py on an MI355X; absent file = defaults."""
    import json
    import os
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "gfx950_routing.json")
    if not os.path.exists(path):
        return
    try:
        with open(path) as f:
            r = json.load(f)
        mod.set_routing(int(r.get("nt_split_budget", -1)),
                        int(r.get("tn_split_budget", -1)),
                        int(r.get("tn2_split_budget", -1)),
                        int(r.get("ntn_max_n", -1)))
        # per-(M,N,K) dense-GEMM split rows (tools/gen_fc_splits.py):
        # the budget formula over-splits small-tile FC shapes.
        # VELES_NO_FC_TABLE=1 skips them (A/B and debugging)
        if os.environ.get("VELES_NO_FC_TABLE", "0") != "1":
            for is_tn, field in ((0, "fc_nt_splits"),
                                 (1, "fc_tn_splits")):
                for row in r.get(field, []):
                    m_, n_, k_, s_ = (int(v) for v in row)
                    mod.set_fc_split(is_tn, m_, n_, k_, s_)
    except Exception as e:                       # never block import
        import logging
        logging.getLogger("ops").warning("routing table ignored: %r", e)


def ext():
    return _load()


def available():
    return _load() is not None


def require():
    if _load() is None:
        raise RuntimeError(
            "the _mzk HIP extension is not built (%s); on a GPU box this "
            "is fatal — run `python veles/znicz_amd/ops/setup.py "
            "build_ext --inplace` (hipcc --offload-arch=gfx950)"
            % _ext_err)
    return _ext


# ---------------------------------------------------------------------------
# tensor helpers
# ---------------------------------------------------------------------------

def as2d_bf16(arr):
    """View an Array's device tensor as 2-D [batch, features], bf16."""
    import torch
    t = arr.devmem
    assert t is not None, "Array has no device tensor (%r)" % (arr,)
    assert t.dtype == torch.bfloat16, \
        "expected bf16 device tensor, got %s" % t.dtype
    return t.view(t.shape[0], -1)


def as2d(arr):
    t = arr.devmem
    return t.view(t.shape[0], -1)


def attach_working_copies(weights_arr, transposed=False):
    """Give a weights Array its bf16 working copy and bf16ᵀ copy.

    weights_arr.devmem is the fp32 master; .bf16 mirrors it; .bf16_t is
    the transposed bf16 image used by whichever of fwd/dgrad needs the
    opposite layout.  The fused SGD kernel refreshes both.
    """
    import torch
    w = weights_arr.devmem
    if getattr(weights_arr, "bf16", None) is None:
        weights_arr.bf16 = w.to(torch.bfloat16)
        weights_arr.bf16_t = w.t().contiguous().to(torch.bfloat16)
    return weights_arr


def refresh_working_copies(weights_arr):
    """Host-side refresh (used after map_write edits); the SGD kernel
    does this on-device in the normal path."""
    w = weights_arr.devmem
    if getattr(weights_arr, "bf16", None) is not None:
        import torch
        weights_arr.bf16.copy_(w.to(torch.bfloat16))
        if getattr(weights_arr, "bf16_t", None) is not None:
            weights_arr.bf16_t.copy_(w.t().to(torch.bfloat16))


def refresh_all_copies(arr):
    """Rebuild every bf16 working copy an Array carries from its fp32
    master ``devmem`` — fc arrays have ``bf16``+``bf16_t``, conv weight
    arrays ``bf16``+``bf16_dgrad``, biases sometimes just ``bf16``.
    Used after host-side weight edits (NNRollback, snapshot surgery)."""
    if arr.devmem is None or getattr(arr, "bf16", None) is None:
        return
    arr.bf16.copy_(arr.devmem.reshape(arr.bf16.shape).to(arr.bf16.dtype))
    if getattr(arr, "bf16_t", None) is not None:
        refresh_fc_copies(arr)
    if getattr(arr, "bf16_dgrad", None) is not None:
        refresh_conv_copies(arr)


def labels_on_device(labels_arr, device):
    """int32 labels tensor on the compute device (cached on the Array)."""
    import torch
    t = getattr(labels_arr, "_dev_i32", None)
    if t is None or t.shape[0] != labels_arr.shape[0]:
        t = torch.empty(labels_arr.shape[0], dtype=torch.int32,
                        device=device.torch_device)
        labels_arr._dev_i32 = t
    if labels_arr.devmem is not None and \
            labels_arr.devmem.dtype == torch.int32:
        return labels_arr.devmem
    import numpy
    t.copy_(torch.from_numpy(numpy.ascontiguousarray(
        labels_arr.mem.astype(numpy.int32))), non_blocking=True)
    return t


# ---------------------------------------------------------------------------
# kernel wrappers (1:1 with bindings.cpp)
# ---------------------------------------------------------------------------

def gemm_nt(a, b, out, bias=None, activation=0, dact=None):
    """out[M,N] = act(a[M,K] · b[N,K]ᵀ + bias).  a,b bf16; out bf16|f32.

    With ``dact`` (bf16, same shape as out) the epilogue instead FUSES
    the consumer layer's activation derivative: out = (a·bᵀ) ∘
    act'(dact) — replaces the separate act_bwd pass on the dgrad path
    (``activation`` is then the derivative code)."""
    require().gemm_nt(a, b, out, bias, int(activation), dact)


def gemm_tn(a, b, out, beta=0.0, bias_out=None):
    """out[M,N] (fp32) = a[K,M]ᵀ · b[K,N] (+beta*out).  a,b bf16.

    With ``bias_out`` (fp32[M]) the kernel co-produces the column sums
    of ``a`` — the bias gradient when a is err_output — for free during
    operand staging.  Returns True iff it did (the vectorized TN2 route
    only); on False the caller must col_sum separately."""
    return require().gemm_tn(a, b, out, float(beta), bias_out)


def act_bwd_inplace(err, output, activation):
    """err ∘= act'(output) — both bf16, same shape."""
    require().act_bwd(err, output, int(activation))


def act_fwd_inplace(x, activation):
    require().act_fwd(x, int(activation))


def col_sum(a, out, accumulate=False):
    """out[N] (fp32) = (out +) Σ_rows a[M,N] (bf16)."""
    require().col_sum(a, out, bool(accumulate))


def sgd_step(param_arr, grad, vel, lr, decay, l1_vs_l2, moment, scale):
    """Fused SGD: master fp32 update + bf16 working-copy refresh.

    grad is the raw batch-summed fp32 gradient; ``scale`` (=1/batch or
    1/(batch*world)) normalizes it inside the kernel.  Extra layout
    copies (bf16ᵀ, conv dgrad image) are refreshed by the owning GD
    unit's ``post_update()``.
    """
    master = param_arr.devmem
    bf16 = getattr(param_arr, "bf16", None)
    require().sgd_step(master.view(-1), grad.view(-1), vel.view(-1),
                       None if bf16 is None else bf16.view(-1), None,
                       float(lr), float(decay), float(l1_vs_l2),
                       float(moment), float(scale))


def sgd_multi_prepare(units, world):
    """Build the one-launch-per-bucket SGD descriptor blobs for the GD
    units of a gradient bucket.  Returns (desc_blob_dev, b2t_dev);
    reusable across steps while hyper-parameters are unchanged (the
    kernel reads only device memory — hipGraph-replay safe)."""
    import torch
    w, g, vel, wb = [], [], [], []
    lr, decay, l1, moment, scale = [], [], [], [], []

    def push(param_arr, grad, v, lr_, decay_, l1_, moment_, batch):
        w.append(param_arr.devmem.view(-1))
        g.append(grad.view(-1))
        vel.append(v.view(-1))
        b = getattr(param_arr, "bf16", None)
        wb.append(None if b is None else b.view(-1))
        lr.append(float(lr_))
        decay.append(float(decay_))
        l1.append(float(l1_))
        moment.append(float(moment_))
        scale.append(1.0 / (batch * world))

    for u, batch in units:
        push(u.weights, u.gradient_weights.devmem,
             u.accumulated_gradient_weights.devmem, u.learning_rate,
             u.weights_decay, u.l1_vs_l2, u.gradient_moment, batch)
        if u.include_bias and u.bias is not None and u.bias:
            push(u.bias, u.gradient_bias.devmem,
                 u.accumulated_gradient_bias.devmem, u.learning_rate_bias,
                 u.weights_decay_bias, u.l1_vs_l2_bias,
                 u.gradient_moment_bias, batch)
    blob, tbl = require().sgd_multi_prepare(w, g, vel, wb, lr, decay, l1,
                                            moment, scale)
    dev = w[0].device
    return blob.to(dev), tbl.to(dev)


def sgd_multi(desc_blob_dev, b2t_dev):
    """Apply the fused SGD update to every tensor of a prepared bucket
    in ONE kernel launch."""
    require().sgd_multi(desc_blob_dev, b2t_dev)


def attach_conv_copies(weights_arr):
    """Conv weights [K,KY,KX,C]: bf16 GEMM image [K, KY·KX·C] and the
    dgrad image [C,KY,KX,K] (= W permuted, no flip — the dgrad loader
    indexes (kh,kw) directly)."""
    import torch
    w = weights_arr.devmem
    if getattr(weights_arr, "bf16", None) is None:
        weights_arr.bf16 = w.reshape(w.shape[0], -1).to(torch.bfloat16) \
            .contiguous()
        weights_arr.bf16_dgrad = w.permute(3, 1, 2, 0).contiguous().to(
            torch.bfloat16)
    return weights_arr


def refresh_conv_copies(weights_arr):
    # [K,KY,KX,C] bf16 working copy -> [C,KY,KX,K] dgrad image, one
    # LDS-tiled HIP transpose (torch's strided copy_ was ~1 TB/s)
    K, KY, KX, C = weights_arr.devmem.shape
    require().permute_kc(weights_arr.bf16.view(-1),
                         weights_arr.bf16_dgrad.view(-1), K, KY * KX, C)


def refresh_fc_copies(weights_arr):
    """bf16ᵀ image from the (just-refreshed) bf16 working copy."""
    require().transpose2d(weights_arr.bf16, weights_arr.bf16_t)


def softmax_fwd(x, max_idx):
    """Row-wise in-place softmax on fp32 x[M,N]; writes argmax to
    max_idx[M] (int32)."""
    require().softmax_fwd(x, max_idx)


def eval_softmax(probs, labels, err_out, n_err, batch):
    require().eval_softmax(probs, labels, err_out, n_err, int(batch))


def confusion_accum(max_idx, labels, cm, batch):
    """cm[pred, label] += 1 on device (int32) — no host sync."""
    require().confusion_accum(max_idx, labels, cm, int(batch))


def histogram(x, counts, lo, hi):
    """counts[nbins] (int32, device) += histogram of x over [lo, hi]
    with numpy.histogram edge semantics (LDS hist kernel)."""
    require().histogram(x, counts, float(lo), float(hi))


def eval_mse(y, target, err_out, mse, metrics, batch, root):
    require().eval_mse(y, target, err_out, mse, metrics, int(batch),
                       int(root))


# conv / pooling / LRN / dropout -------------------------------------------

def conv_fwd(x, w, out, bias, conv_desc, activation=0):
    require().conv_fwd(x, w, out, bias, conv_desc, int(activation))


def conv_dgrad(err_out, w_dgrad, err_in, conv_desc, activation=0,
               dact=None):
    """err_in = col2im-gather(err_out · Wᵀ); with ``dact`` the consumer
    layer's activation derivative is fused into the epilogue (see
    gemm_nt)."""
    require().conv_dgrad(err_out, w_dgrad, err_in, conv_desc,
                         int(activation), dact)


def conv_wgrad(err_out, x, grad_w, conv_desc, beta=0.0, bias_out=None):
    """Returns True iff bias_out (fp32[K]) was co-produced (see
    gemm_tn)."""
    return require().conv_wgrad(err_out, x, grad_w, conv_desc,
                                float(beta), bias_out)


def pool_fwd(x, out, offs, desc, mode):
    require().pool_fwd(x, out, offs, desc, int(mode))


def pool_bwd(err_out, err_in, offs, desc, mode):
    require().pool_bwd(err_out, err_in, offs, desc, int(mode))


def lrn_fwd(x, out, denom, alpha, beta, k, n):
    """denom (fp32, optional): pass None on the training path — the
    backward kernel recomputes the denominators from x."""
    require().lrn_fwd(x, out, denom, float(alpha), float(beta), float(k),
                      int(n))


def lrn_bwd(err_out, x, out, denom, err_in, alpha, beta, k, n):
    require().lrn_bwd(err_out, x, out, denom, err_in, float(alpha),
                      float(beta), float(k), int(n))


def dropout_fwd(x, out, mask, ratio, seed, counter, ctr_buf=None):
    """ctr_buf (int64[1] device tensor): device-resident step counter,
    bumped by a leading kernel — makes the launch hipGraph-replayable."""
    require().dropout_fwd(x, out, mask, float(ratio), int(seed),
                          int(counter), ctr_buf)


def dropout_bwd(err, mask, ratio):
    """err ∘= mask * 1/(1-ratio), in place."""
    require().dropout_bwd(err, mask, float(ratio))


def transpose2d(a, out):
    require().transpose2d(a, out)


# misc (depooling / RBM / Kohonen / input-norm / Uniform) -------------------

def gather_by_offset(src, offs, dst):
    """dst[i] = src[offs[i]] (bf16, int32 absolute offsets)."""
    require().gather_by_offset(src, offs, dst)


def depool_scatter(src, offs_u8, dst, pool_desc):
    """Unpooling forward over u8 relative winner codes: dst (large,
    zeroed by the kernel) gets src at each stored window position."""
    require().depool_scatter(src, offs_u8, dst, pool_desc)


def depool_gather(src, offs_u8, dst, pool_desc):
    """Unpooling backward: dst[o] = src[winner(o)] (0 when none)."""
    require().depool_gather(src, offs_u8, dst, pool_desc)


def scatter_by_offset(src, offs, dst):
    """dst zeroed then dst[offs[i]] = src[i] (unique offsets)."""
    require().scatter_by_offset(src, offs, dst)


def bernoulli(x, y, seed):
    """y = (U < x) ? 1 : 0, counter-based PRNG (bf16 in/out)."""
    require().bernoulli(x, y, int(seed))


def fill_uniform(y, lo, hi, seed):
    """fp32 uniform fill in [lo, hi) (reference Uniform unit)."""
    require().fill_uniform(y, float(lo), float(hi), int(seed))


def row_argmin(d, winners):
    """winners[r] = argmin_c d[r, c] (fp32 rows, int32 out)."""
    require().row_argmin(d, winners)


def mean_disp(x, mean, rdisp, y):
    """y = (x - mean) * rdisp, mean/rdisp broadcast over batch."""
    require().mean_disp(x, mean, rdisp, y)
