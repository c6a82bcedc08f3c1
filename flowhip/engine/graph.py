"""hipGraph-captured inference (BASELINE config 4).

Captures the fixed-shape iterative-refinement forward into one hipGraph
(torch.cuda.CUDAGraph is hipGraph on ROCm) and replays it per frame pair —
removing the per-iteration launch overhead of the 24-32 GRU iterations.

Capture-safety of the model forward is by construction: the corr-lookup
window offsets are device-resident compile-time constants (no host->device
delta transfer — SURVEY.md §2.9 quirk 8), coords grids are created on
device, and no host syncs occur in the loop.
"""

import torch


class GraphedInference:
    """Wrap a model for fixed-shape graph-replayed test_mode inference.

    Usage:
        g = GraphedInference(model, (1, 3, 288, 960), iters=24)
        flow_low, flow_up = g(image1, image2)
    Inputs must match the capture shape; outputs are views of static buffers
    (clone them to retain across calls).
    """

    def __init__(self, model, input_shape, iters, warmup=3):
        assert torch.cuda.is_available(), "hipGraph capture requires a GPU"
        self.model = model.eval()
        self.iters = iters
        device = next(model.parameters()).device

        self.static_img1 = torch.zeros(input_shape, device=device)
        self.static_img2 = torch.zeros(input_shape, device=device)

        # warm up on a side stream (allocator + MIOpen autotune settle)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(warmup):
                self.model(self.static_img1, self.static_img2,
                           iters=iters, test_mode=True)
        torch.cuda.current_stream().wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph), torch.no_grad():
            self.static_out = self.model(self.static_img1, self.static_img2,
                                         iters=iters, test_mode=True)

    @torch.no_grad()
    def __call__(self, image1, image2):
        self.static_img1.copy_(image1)
        self.static_img2.copy_(image2)
        self.graph.replay()
        return self.static_out
