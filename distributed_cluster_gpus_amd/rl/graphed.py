"""hipGraph-captured SAC train step.

The CHSAC-AF networks are tiny (~0.6 M params); one train step issues ~200
small kernels, so on MI355X the step is launch-bound (~7 ms, ~140 steps/s)
rather than compute-bound.  Per the CDNA4 playbook ("capture launch-bound
inner loops in hipGraphs"), this wraps sample+update into one captured graph:
after capture, each step is a single graph replay plus one 2 KB H2D index
copy.

Requirements/limits:
* the agent must be built with ``graph_capturable=True`` (Adam capturable
  state) — the batched engine does this automatically on GPU;
* replay storage is the capture-static memory: ReplayRing's preallocated
  tensors are gathered by a static device index tensor, so new transitions
  written between replays are picked up naturally;
* not combined with the DP all-reduce hook (RCCL capture support varies);
  multi-rank training uses the eager path;
* falls back transparently to the eager step if capture fails.
"""
from typing import Optional

import torch


class GraphedSACStep:
    def __init__(self, agent, replay, batch_size: int, warmup_iters: int = 3):
        self.agent = agent
        self.replay = replay
        self.batch_size = int(batch_size)
        self.device = agent.device
        assert self.device.type == "cuda", "graph capture needs a GPU"
        assert agent.algo.allreduce_hook is None, \
            "graphed step does not support the DP all-reduce hook"
        self.idx = torch.zeros(self.batch_size, dtype=torch.long,
                               device=self.device)
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.capture_error: Optional[str] = None
        self._capture(warmup_iters)

    def _batch_from_idx(self):
        r = self.replay
        return {
            "s": r.s[self.idx], "s_next": r.s_next[self.idx],
            "a_dc": r.a_dc[self.idx], "a_g": r.a_g[self.idx],
            "r": r.r[self.idx], "done": r.done[self.idx],
            "costs": {name: r.costs[self.idx, k]
                      for k, name in enumerate(r.cost_names)},
            "mask_dc": r.mask_dc[self.idx], "mask_g": r.mask_g[self.idx],
        }

    def _eager(self):
        self.agent.train_step(self._batch_from_idx(), compute_stats=False)

    def _fill_idx(self):
        """Sample indices ON-DEVICE (a small randint kernel).  A host-side
        randint + pageable H2D copy_ here serializes against every BLOCKING
        stream via legacy-default-stream semantics — with the advance kernel
        on the CU-masked (blocking) stream, each train step would wait out
        the whole simulation window (measured: exactly one overlapped step
        per cycle, at any island size / grid shape)."""
        n = max(1, self.replay.size)
        torch.randint(0, n, (self.batch_size,), device=self.device,
                      out=self.idx)

    def _capture(self, warmup_iters: int):
        self._fill_idx()
        try:
            side = torch.cuda.Stream(device=self.device)
            side.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(side):
                for _ in range(warmup_iters):
                    self._eager()
            torch.cuda.current_stream(self.device).wait_stream(side)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._eager()
            self.graph = g
        except Exception as e:  # pragma: no cover - GPU only
            self.capture_error = repr(e)
            self.graph = None  # eager fallback

    @property
    def captured(self) -> bool:
        return self.graph is not None

    def step(self):
        self._fill_idx()
        if self.graph is not None:
            self.graph.replay()
        else:
            self._eager()
