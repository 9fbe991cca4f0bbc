"""torch_actor_critic_amd — an MI355X-native Soft Actor-Critic training framework.

Brand-new implementation with the capabilities, public API surface, and
checkpoint layout of ``dogeplusplus/torch-actor-critic`` (the reference),
re-designed for AMD Instinct MI355X (gfx950):

* MLP / convolutional actor-critic networks whose hot path runs on
  hand-written CDNA4 HIP kernels (MFMA GEMM, fused tanh-Gaussian head,
  fused Bellman backup) — see ``torch_actor_critic_amd.ops``.
* A replay buffer resident in HBM3E with a fused on-device Philox
  uniform-sample + gather kernel — see ``torch_actor_critic_amd.buffer``.
* Flat-buffer parameter/optimizer storage so the polyak target update and
  the Adam step are each ONE kernel launch and the data-parallel gradient
  reduction is ONE RCCL all-reduce over xGMI — see
  ``torch_actor_critic_amd.parallel`` and ``torch_actor_critic_amd.optim``.
* The whole SAC update step captured into a hipGraph and replayed.

Top-level compat packages ``sac``, ``networks``, ``buffer``,
``environments`` re-export this package's classes under the reference's
module paths (reference: main.py:12-20).
"""

__version__ = "0.1.0"
