"""Error types for the veles MI355X-native framework.

Parity with reference veles/error.py (error classes used across the
framework).
"""


class VelesException(Exception):
    """Base class for all framework errors."""


class BadFormatError(VelesException):
    """Raised when data has an unexpected format or shape."""


class NotExistsError(VelesException):
    """Raised when a required attribute/entity does not exist."""


class AlreadyExistsError(VelesException):
    """Raised when an entity that must be unique already exists."""


class DeviceError(VelesException):
    """Raised on device (HIP/ROCm) failures."""


class MasterSlaveCommunicationError(VelesException):
    """Kept for API parity; the MI355X engine uses RCCL collectives, so
    this maps to collective-communication failures."""
