"""Package metadata.

Reference parity: /root/reference/src/service/metadata.py:10 (__version__).
"""

__version__ = "0.1.0"
__framework__ = "detectmate-mi355x"
