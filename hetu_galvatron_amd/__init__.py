"""hetu_galvatron_amd — MI355X-native automatic hybrid-parallel training framework.

A from-scratch rebuild of the capabilities of PKU-DAIR/Hetu-Galvatron for AMD
Instinct MI355X (gfx950, CDNA4): PyTorch-ROCm + hand-written HIP kernels + RCCL
over xGMI.  See SURVEY.md at the repo root for the capability blueprint and
file:line citations into the reference.

Layers:
  config/    - YAML + dotted-override loading into pydantic schemas; the
               per-layer strategy JSON codec (the search<->runtime contract).
  core/      - distributed initialization, global state, communication-group
               fabric (rank coordinates over pp-dp-cp-tp-sp).
  runtime/   - tensor parallel layers, transformer modules, the ZeRO engine
               (flat-param blocks with explicit grad-sync hooks, no FSDP
               monkey-patching), pipeline schedules (GPipe / 1F1B), optimizer,
               synthetic + real dataloaders, checkpointing.
  ops/       - HIP/CDNA4 kernels (rmsnorm, rope, swiglu, fused adam,
               vocab-parallel cross-entropy, flash attention w/ LSE) with
               CPU-only torch fallbacks for GPU-less test runs.
  search/    - analytic cost models + dynamic-programming search engine
               (C++ pybind11 DP core).
  profiler/  - hardware (RCCL/xGMI bandwidth sweeps), model (per-layer
               time/memory), and runtime (iteration timing, peak memory)
               profilers.
"""

__version__ = "0.1.0"

from .config import GalvatronConfig, load_config  # noqa: F401
