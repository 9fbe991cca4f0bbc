"""Fast environment-interaction path.

The env loop is host-side and serial (MuJoCo-like physics can't move to
the GPU — SURVEY.md §7 hard parts), so per-step cost is pure latency:
an eager B=1 actor forward is ~15 Python-dispatched kernel launches plus
a device sync.  This module cuts it to: one pinned H2D copy, ONE
hipGraph replay (the whole actor forward incl. Philox noise), one pinned
D2H copy, one event sync (~25 us/step instead of ~1 ms).

``WindowedStore`` batches the interaction window's transitions in host
pinned buffers and flushes them to the HBM ring ONCE per update window —
semantically identical to per-step stores (the reference samples only at
burst time, sac/algorithm.py:273-278) but 6 launches per window instead
of 6 per step.
"""

import numpy as np
import torch


class ActKernel:
    """Single-kernel acting path: the whole stochastic actor forward for
    one state runs as ONE launch (ops/csrc/fused.hip::act_kernel) — no
    graph replay, no per-layer launches, self-bumped Philox noise."""

    def __init__(self, actor, obs_dim: int, act_dim: int,
                 device: torch.device, philox_seed: int = 0):
        from ..models.mlp import Actor as MlpActor
        from ..ops import require_extension
        if type(actor) is not MlpActor:
            raise TypeError("ActKernel needs the plain MLP Actor")
        if any(l.out_features > 256 for l in actor.layers) or act_dim > 64:
            raise ValueError("shape outside act-kernel limits")
        self.ext = require_extension()
        self.ws = [l.weight for l in actor.layers]
        self.bs = [l.bias for l in actor.layers]
        self.wmu = actor.mu_layer.weight
        self.bmu = actor.mu_layer.bias
        self.wls = actor.log_std_layer.weight
        self.bls = actor.log_std_layer.bias
        self.act_limit = float(actor.act_limit)
        self.lo = float(actor.log_min_std)
        self.hi = float(actor.log_max_std)
        self.seed = philox_seed ^ 0x5ACB
        self.ctr = torch.zeros(1, dtype=torch.int64, device=device)
        self.obs_in = torch.zeros(obs_dim, device=device)
        self.obs_pin = torch.zeros(obs_dim, pin_memory=True)
        self.act_out = torch.zeros(act_dim, device=device)
        self.act_pin = torch.zeros(act_dim, pin_memory=True)
        self.flag_pin = torch.zeros(1, dtype=torch.int32, pin_memory=True)
        self.flag_np = self.flag_pin.numpy()
        self.ev = torch.cuda.Event()
        import os
        self._pinned_ok = os.environ.get("TAC_AMD_ACT_PINNED", "1") != "0"
        # pre-marshalled launch handle (referenced tensors stay alive via
        # self.*): one int-arg pybind call per env step
        self._handle = self.ext.act_prepare(
            self.obs_pin, self.ws, self.bs, self.wmu, self.bmu, self.wls,
            self.bls, self.act_out, self.act_pin, self.flag_pin, self.ctr,
            self.seed, self.act_limit, self.lo, self.hi)

    def act(self, state: np.ndarray) -> np.ndarray:
        self.obs_pin.copy_(torch.from_numpy(np.asarray(state,
                                                       dtype=np.float32)))
        if self._pinned_ok:
            # zero-copy I/O: the kernel READS the observation from and
            # WRITES the action + a system-scope completion flag to
            # host-pinned memory; the host spins on the flag — no H2D/D2H
            # copies, no event sync
            self.flag_pin[0] = 0
            self.ext.act_fire(self._handle)
            fl = self.flag_np
            for _ in range(2_000_000):
                if fl[0] != 0:
                    return self.act_pin.numpy().copy()
            # flag never landed: drain the stream once, then give up on
            # the pinned path for this session
            torch.cuda.synchronize()
            if fl[0] != 0:
                return self.act_pin.numpy().copy()
            self._pinned_ok = False
        self.obs_in.copy_(self.obs_pin, non_blocking=True)
        self.ext.act_step(self.obs_in, self.ws, self.bs, self.wmu,
                          self.bmu, self.wls, self.bls, self.act_out,
                          self.ctr, self.seed, self.act_limit, self.lo,
                          self.hi)
        self.act_pin.copy_(self.act_out, non_blocking=True)
        self.ev.record()
        self.ev.synchronize()
        return self.act_pin.numpy().copy()


def make_act_path(actor, obs_dim, act_dim, device, philox_seed: int = 0):
    """Best available single-state acting path: the one-launch act
    kernel when the actor fits its limits, else the captured act-graph."""
    try:
        return ActKernel(actor, obs_dim, act_dim, device, philox_seed)
    except Exception:  # noqa: BLE001
        return ActGraph(actor, obs_dim, act_dim, device)


class ActGraph:
    """hipGraph-captured single-state stochastic actor forward."""

    def __init__(self, actor, obs_dim: int, act_dim: int,
                 device: torch.device, warmup: int = 3):
        self.device = device
        self.obs_in = torch.zeros(1, obs_dim, device=device)
        self.obs_pin = torch.zeros(obs_dim, pin_memory=True)
        self.act_pin = torch.zeros(1, act_dim, pin_memory=True)
        self.ev = torch.cuda.Event()

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(warmup):
                actor(self.obs_in, deterministic=False, with_logprob=False)
        torch.cuda.current_stream().wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph), torch.no_grad():
            self.act_out, _ = actor(self.obs_in, deterministic=False,
                                    with_logprob=False)

    def act(self, state: np.ndarray) -> np.ndarray:
        self.obs_pin.copy_(torch.from_numpy(np.asarray(state,
                                                       dtype=np.float32)))
        self.obs_in.copy_(self.obs_pin.view(1, -1), non_blocking=True)
        self.graph.replay()
        self.act_pin.copy_(self.act_out, non_blocking=True)
        self.ev.record()
        self.ev.synchronize()
        return self.act_pin[0].numpy().copy()


class WindowedStore:
    """Host-side staging of up to `window` transitions, flushed to the
    replay ring in one batched write."""

    def __init__(self, buffer, window: int):
        self.buffer = buffer
        obs_dim, act_dim = buffer.obs_dim, buffer.act_dim
        self.obs = np.zeros((window, obs_dim), dtype=np.float32)
        self.act = np.zeros((window, act_dim), dtype=np.float32)
        self.rew = np.zeros(window, dtype=np.float32)
        self.nobs = np.zeros((window, obs_dim), dtype=np.float32)
        self.done = np.zeros(window, dtype=np.float32)
        self.n = 0
        self.window = window

    def store(self, obs, act, rew, next_obs, done):
        i = self.n
        self.obs[i] = obs
        self.act[i] = act
        self.rew[i] = rew
        self.nobs[i] = next_obs
        self.done[i] = done
        self.n += 1
        if self.n == self.window:
            self.flush()

    def flush(self):
        if self.n == 0:
            return
        n = self.n
        self.buffer.store_batch(self.obs[:n], self.act[:n], self.rew[:n],
                                self.nobs[:n], self.done[:n])
        self.n = 0


class VisualActGraph:
    """hipGraph-captured single-state stochastic VISUAL actor forward
    (conv trunk + dual-stream tanh-Gaussian head) with pinned-staging
    I/O — replaces ~0.4 ms of eager per-step acting with two H2D
    copies + one graph replay + one D2H copy.  Philox noise is drawn by
    a captured device-counter kernel, so every replay acts with fresh
    noise (same mechanism as the update graphs)."""

    def __init__(self, actor, feat_dim: int, vis_dim, act_dim: int,
                 device: torch.device, warmup: int = 3):
        from ..envs.visual import MultiObservation
        self.device = device
        self.feat_in = torch.zeros(feat_dim, device=device)
        self.frame_in = torch.zeros(*vis_dim, device=device)
        self.feat_pin = torch.zeros(feat_dim, pin_memory=True)
        self.frame_pin = torch.zeros(*vis_dim, pin_memory=True)
        self.act_pin = torch.zeros(act_dim, pin_memory=True)
        self.ev = torch.cuda.Event()
        mo = MultiObservation(self.feat_in, self.frame_in)

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(warmup):
                actor(mo, deterministic=False, with_logprob=False)
        torch.cuda.current_stream().wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph), torch.no_grad():
            self.act_out, _ = actor(mo, deterministic=False,
                                    with_logprob=False)

    def act(self, state, buffer=None) -> np.ndarray:
        # when the buffer's fused store already staged this observation
        # on device (state is last step's next_obs), read it D2D instead
        # of re-uploading the 84 KB frame through pinned memory
        staged = None
        if buffer is not None:
            last = getattr(buffer, "_last_next_src", None)
            st = getattr(buffer, "_stage", None)
            if (last is not None and st is not None
                    and state.features is last[0]
                    and state.frame is last[1]):
                staged = (st["nf"], st["nF"])
        if staged is not None:
            self.feat_in.copy_(staged[0], non_blocking=True)
            self.frame_in.copy_(staged[1], non_blocking=True)
        else:
            self.feat_pin.copy_(state.features.reshape(-1))
            self.frame_pin.copy_(state.frame)
            self.feat_in.copy_(self.feat_pin, non_blocking=True)
            self.frame_in.copy_(self.frame_pin, non_blocking=True)
        self.graph.replay()
        self.act_pin.copy_(self.act_out.reshape(-1), non_blocking=True)
        self.ev.record()
        self.ev.synchronize()
        return self.act_pin.numpy().copy()
