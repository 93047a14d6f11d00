"""hipGraph-captured training step.

The Evoformer step launches thousands of kernels (forward + per-block
checkpoint recompute + backward + optimizer); at dim=256/depth=12 the
Python dispatch path costs ~260 ms/step on MI355X while the GPU work is
~300 ms — the step is launch-bound.  Capturing one full training step
(zero_grad -> forward -> backward -> optimizer) into a hipGraph and
replaying it removes the host path entirely (the CDNA guide's "capture
launch-bound inner loops in hipGraphs").

Requirements on the step function:
* static input tensors (copy new data into them between replays),
* optimizer constructed with capturable=True (Adam/AdamW),
* gradients materialized before capture (warmup steps) so buffer
  addresses are stable; zero_grad(set_to_none=False) inside the step.

torch.cuda.CUDAGraph IS hipGraph on ROCm.
"""
import torch


class GraphedTrainStep:
    """Capture `step_fn` (a no-arg callable returning a scalar loss
    tensor) into a hipGraph after `warmup` eager runs.

    On capture failure (unsupported op, collective restrictions) falls
    back to eager execution transparently — check `.graphed`.
    """

    def __init__(self, step_fn, warmup=3, fallback=True):
        self.step_fn = step_fn
        self.graph = None
        self.static_loss = None
        self.graphed = False

        assert torch.cuda.is_available(), "GraphedTrainStep needs a device"

        # warmup on a side stream so allocations stabilize
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(max(1, warmup)):
                step_fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        self.capture_error = None
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self.static_loss = step_fn()
            self.graph = g
            self.graphed = True
        except Exception as e:
            if not fallback:
                raise
            self.capture_error = e
            self.graph = None
            self.graphed = False
            torch.cuda.synchronize()

    def __call__(self):
        if self.graphed:
            self.graph.replay()
            return self.static_loss
        return self.step_fn()
