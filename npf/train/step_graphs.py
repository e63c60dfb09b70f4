"""Shape-keyed hipGraph capture of whole optimization steps.

Eager NPF steps are launch-bound on MI355X: an AttnCNP-1D step is ~300
kernels of a few µs each, so replaying the step as ONE hipGraph is a ~10x
wall-clock lever (bench.py measured 1.43 ms graphed vs ~35 ms through the
eager trainer loop).  Training episodes have a VARIABLE context count
(GetRandomIndcs draws 0..50), so this module keeps one captured graph per
episode shape: the splitter runs outside the graphs, episodes are copied
into the shape's static buffers, and the graph replays forward + backward +
flat-buffer reduce + fused-Adam step.

Fidelity guarantees:
- capture warmup steps run on a saved copy of ALL mutable state (params,
  Adam state, BN running stats) which is restored before the first real
  replay — warmup does not consume optimization steps;
- latent sampling inside a graph goes through the static noise pool
  (npf.ops.noise), refreshed outside the graph every step;
- the LR schedule writes into the capturable optimizer's device lr tensor,
  which every replay reads.
"""

import gc
import logging

import torch

logger = logging.getLogger(__name__)

__all__ = ["GraphedStepper"]


def _flat_key(inputs, y):
    parts = []
    for k in sorted(inputs):
        v = inputs[k]
        parts.append((k, tuple(v.shape) if torch.is_tensor(v) else v))
    parts.append(("__y", tuple(y.shape)))
    return tuple(parts)


class GraphedStepper:
    """Run `step_fn(inputs, y) -> loss` through per-shape captured graphs.

    step_fn must be replay-safe: fixed tensor addresses for parameters,
    grads and optimizer state; any RNG routed through the noise pool.
    """

    def __init__(self, module, optimizer, step_fn, max_graphs=80, warmup=3):
        self.module = module
        self.optimizer = optimizer
        self.step_fn = step_fn
        self.max_graphs = max_graphs
        self.warmup = warmup
        self.graphs = {}
        self.disabled = False

    # -------------------------- state snapshot -------------------------- #

    def _mutable_tensors(self):
        ts = list(self.module.parameters())
        ts += [b for b in self.module.buffers() if b.is_floating_point()
               or b.dtype in (torch.int64, torch.int32)]
        for group in self.optimizer.param_groups:
            for p in group["params"]:
                st = self.optimizer.state.get(p, {})
                ts += [v for v in st.values() if torch.is_tensor(v)]
        return ts

    def _save_state(self):
        return [t.detach().clone() for t in self._mutable_tensors()]

    def _restore_state(self, saved):
        for t, s in zip(self._mutable_tensors(), saved):
            t.detach().copy_(s)

    # ----------------------------- capture ------------------------------ #

    def _capture(self, inputs, y):
        from npf.ops import noise

        static_in = {k: v.clone() for k, v in inputs.items()}
        static_y = y.clone()

        # warmup on a side stream primes allocator blocks, BLAS algo caches,
        # optimizer state tensors and the noise-pool buffers for this shape.
        # The pool is enabled ONLY here (capture bakes the pool read; eager
        # eval keeps plain rsample semantics).
        saved = self._save_state()
        was_enabled = noise.is_noise_pool_enabled()
        noise.enable_noise_pool(True)
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(self.warmup):
                    noise.refresh_noise_()
                    wl = self.step_fn(static_in, static_y)
            del wl
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            # warmup-created autograd nodes must be released before capture
            gc.collect()

            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                static_loss = self.step_fn(static_in, static_y)
        finally:
            noise.enable_noise_pool(was_enabled)
        # warmup must not consume optimization steps
        self._restore_state(saved)
        return dict(graph=g, inputs=static_in, y=static_y, loss=static_loss)

    # ------------------------------ step -------------------------------- #

    def step(self, inputs, y):
        """Replay (capturing on first sight of a shape); returns the loss
        tensor, or None if graphs are disabled/over budget for this shape."""
        if self.disabled:
            return None
        from npf.ops import noise

        key = _flat_key(inputs, y)
        entry = self.graphs.get(key)
        if entry is None:
            if len(self.graphs) >= self.max_graphs:
                return None
            try:
                entry = self._capture(inputs, y)
            except Exception as e:
                logger.warning("hipGraph capture failed (%r); running eager", e)
                self.disabled = True
                return None
            self.graphs[key] = entry
        for k, v in inputs.items():
            entry["inputs"][k].copy_(v, non_blocking=True)
        entry["y"].copy_(y, non_blocking=True)
        noise.refresh_noise_()
        entry["graph"].replay()
        return entry["loss"]
