"""harmony_amd — MI355X-native (gfx950/CDNA4) rebuild of harmony-one/harmony's
BLS12-381 signature aggregation + pairing-verification hot path.

Layers:
  harmony_amd.core    — ctypes over the C-ABI (include/hbls.h) of libhbls.so,
                        the HIP kernel library.  THE PRODUCT PATH (GPU only).
  harmony_amd.bls     — mirror of the Go crypto/bls surface (SecretKey,
                        PublicKey, Sign, Mask, AggregateSig ...).
  harmony_amd.quorum  — mirror of consensus/quorum decider host logic around
                        the batch GPU API.

Build the native library with `python -m harmony_amd.build`.
"""
__version__ = "0.1"
