"""distrl_llm_amd.ops — HIP/CDNA4 kernel library + pure-torch references.

- ``reference``: plain PyTorch implementations (fp32 ground truth for kernel
  numerics tests, and the CPU execution path).
- ``functional``: dispatch layer — on CUDA (ROCm) tensors it REQUIRES the
  compiled gfx950 extension and fails loudly if missing; on CPU it runs the
  reference implementations.
- ``_C``: the compiled extension (built in-tree, see ``build.py``).
"""

from . import reference  # noqa: F401
