"""kukeon_amd — an MI355X-native self-hosted runtime for AI coding agents.

Two planes, built from scratch (capabilities of eminwux/kukeon, re-designed
MI355X-first — see SURVEY.md):

* control plane: ``kuke`` CLI + ``kukeond`` daemon, Realm/Space/Stack/Cell/
  Session manifests, flock+CAS metadata tree, process-cell runtime with
  amdgpu device pinning (``kukeon_amd.api/ state/ runtime/ controller/
  daemon/ cli/ tty/``).
* data plane: PyTorch-ROCm inference engine with hand-written gfx950 HIP
  kernels, paged KV cache, continuous batching, RCCL tensor parallelism
  (``kukeon_amd.ops/ models/ engine/ parallel/ serve/``).
"""

__version__ = "0.1.0"
