"""Version of the torchsnapshot_amd framework.

Stamped into every snapshot's metadata so readers can check compatibility
(parity with reference torchsnapshot/version.py:19).
"""

__version__ = "0.1.0"

# Lowest metadata version this build can read.
OLDEST_READABLE_VERSION = "0.1.0"
