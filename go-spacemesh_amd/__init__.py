"""go-spacemesh_amd — MI355X-native POST engine (HIP/CDNA4, gfx950).

The product path of this build: a from-scratch reimplementation of the
engine go-spacemesh reaches through the external post-rs library
(activation/post.go PostSetupProvider, activation/post_verifier.go
PostVerifier — see SURVEY.md §8).  The compute lives in libpost_hip.so
(C-ABI, include/spacemesh_post.h); this package is the host-side mirror of
the reference's Go interfaces plus ctypes bindings.

There is NO CPU fallback here: on a machine with a GPU the HIP engine must
load and run, and missing kernels raise immediately.  The CPU oracle under
oracle/ is test infrastructure only and is never imported by this package.
"""
from . import events  # noqa: F401
from .verifier_pool import (  # noqa: F401
    BatchingVerifier,
    OffloadingVerifier,
)
from .api import (  # noqa: F401
    Engine,
    EngineError,
    PostConfig,
    PostProof,
    PostProofMetadata,
    PostSetupManager,
    PostSetupOpts,
    PostVerifier,
    ProveOpts,
    VerifyOpts,
    load_engine,
)

__all__ = [
    "BatchingVerifier", "Engine", "EngineError", "OffloadingVerifier", "PostConfig", "PostProof",
    "PostProofMetadata", "PostSetupManager", "PostSetupOpts", "PostVerifier",
    "ProveOpts", "VerifyOpts", "events", "load_engine",
]
