"""distributedes_amd — an MI355X-native distributed Evolution Strategies framework.

A from-scratch rebuild of the capabilities of ShangtongZhang/DistributedES
(OpenAI-ES + CEM population training), redesigned for AMD Instinct MI355X
(gfx950): vectorized on-GPU populations perturbed from a shared HBM-resident
noise table, hand-written CDNA4 HIP kernels for the hot path (perturb, policy
forward, rank shaping, weighted-sum update, batched env stepping), and RCCL
over xGMI for the per-generation fitness all-gather + parameter all-reduce.

NOTE ON REFERENCE CITATIONS: the reference mount /root/reference was empty in
both the survey and build sessions, so reference file:line
citations are impossible.  The binding spec is BASELINE.json + SURVEY.md; the
reconstructed reference API (run(config)/train_step(state), checkpoint layout
{'network','optimizer','normalizer','generation','rng'}) is frozen per
SURVEY.md §2.9.
"""

__version__ = "0.1.0"

from .config import Config, preset, PRESETS  # noqa: F401
from .api import run, train_step, evaluate  # noqa: F401
