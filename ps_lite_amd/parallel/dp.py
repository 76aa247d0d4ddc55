"""Data-parallel gradient synchronization through the parameter server
(the BytePS training pattern, BASELINE config #4).

Each worker pushes its gradient buckets (tensors split into
<= partition_bytes chunks, the reference's BYTEPS_PARTITION_BYTES
mechanism) and pulls back the cross-worker reduction, then scales by
1/num_workers.

Two server modes:
  * "reduce" (GPU, recommended): servers run the dense handler in
    reduce mode — pulls are held until every worker's push of the round
    arrived, so NO barrier is needed and pushes/pulls of all buckets
    overlap fully.
  * "sum" (CPU demo / default handler): the store accumulates forever,
    so keys rotate every step and a worker barrier separates the push
    and pull phases.
"""

import numpy as np


class PSGradSync:
    def __init__(self, ps, worker, params, num_workers, device=-1,
                 partition_bytes=4 << 20, mode=None, key_base=1 << 20):
        self.ps = ps
        self.worker = worker
        self.params = list(params)
        self.num_workers = num_workers
        self.device = device
        self.mode = mode or ("reduce" if device >= 0 else "sum")
        self.key_base = key_base
        self.step_count = 0

        # bucket layout: (param_idx, elem_offset, elems)
        self.buckets = []
        max_elems = partition_bytes // 4
        for pi, p in enumerate(self.params):
            n = p.numel()
            off = 0
            while off < n:
                take = min(max_elems, n - off)
                self.buckets.append((pi, off, take))
                off += take
        self.bufs = []
        self.pull_bufs = []
        # direct mode (decided on first allreduce): when .grad tensors
        # are pool-resident (ps.use_torch_pool_allocator()), push/pull
        # straight from grad memory — no staging buffers, no copies
        self.direct = None
        for _, _, elems in self.buckets:
            if device >= 0:
                self.bufs.append(ps.pool_alloc(elems * 4))
                self.pull_bufs.append(ps.pool_alloc(elems * 4))
            else:
                self.bufs.append(np.zeros(elems, dtype=np.float32))
                self.pull_bufs.append(np.zeros(elems, dtype=np.float32))

    def _keys(self):
        if self.mode == "reduce":
            base = self.key_base
        else:
            base = self.key_base + self.step_count * len(self.buckets)
        return [base + i for i in range(len(self.buckets))]

    def _ptr(self, buf):
        return buf.ptr if self.device >= 0 else buf.ctypes.data

    def _grad_view(self, b):
        pi, off, elems = self.buckets[b]
        return self.params[pi].grad.detach().reshape(-1)[off:off + elems]

    def allreduce(self):
        """Average .grad across workers, in place. Call after backward."""
        import torch

        if self.direct is None:
            # direct (zero-staging) mode needs every grad bucket inside
            # the zero-copy pool window AND reduce-mode servers (pulls
            # write grads in place)
            self.direct = (self.device >= 0 and self.mode == "reduce" and
                           all(self.ps._core.pool_contains(self._grad_view(b).data_ptr())
                               for b in range(len(self.buckets))))
        keys = self._keys()
        # stage grads into the push buffers (skipped in direct mode)
        if not self.direct:
            for b, (pi, off, elems) in enumerate(self.buckets):
                g = self._grad_view(b)
                if self.device >= 0:
                    # pool buffer <- device grad (same GPU)
                    self.ps._core.k_dense_assign(self.bufs[b].ptr, g.data_ptr(), elems * 4)
                else:
                    self.bufs[b][:] = g.cpu().numpy()
        tss = []
        for b, k in enumerate(keys):
            ka = np.array([k], dtype=np.uint64)
            elems = self.buckets[b][2]
            lens = np.array([elems], dtype=np.int32)
            if self.direct:
                gptr = self._grad_view(b).data_ptr()
                tss.append(self.worker.zpush_ptr(ka, gptr, elems * 4, self.device, lens,
                                                 cmd=2))
                # the reduction is written straight back into .grad
                tss.append(self.worker.zpull_ptr(ka, gptr, elems * 4, self.device, lens))
                continue
            tss.append(self.worker.zpush_ptr(ka, self._ptr(self.bufs[b]), elems * 4,
                                             self.device, lens, cmd=2))
            if self.mode == "reduce":
                tss.append(self.worker.zpull_ptr(ka, self._ptr(self.pull_bufs[b]), elems * 4,
                                                 self.device, lens))
        if self.mode != "reduce":
            for ts in tss:
                self.worker.wait(ts)
            tss = []
            self.ps.barrier("worker", self.ps.WORKER_GROUP)
            for b, k in enumerate(keys):
                ka = np.array([k], dtype=np.uint64)
                elems = self.buckets[b][2]
                lens = np.array([elems], dtype=np.int32)
                tss.append(self.worker.zpull_ptr(ka, self._ptr(self.pull_bufs[b]), elems * 4,
                                                 self.device, lens))
        for ts in tss:
            self.worker.wait(ts)
        # scale to the average (direct mode: the reduction already sits
        # in .grad — the pull wrote it in place over xGMI)
        inv = 1.0 / self.num_workers
        for b, (pi, off, elems) in enumerate(self.buckets):
            g = self._grad_view(b)
            if self.direct:
                g.mul_(inv)
            elif self.device >= 0:
                self.ps._core.k_dense_assign(g.data_ptr(), self.pull_bufs[b].ptr, elems * 4)
                g.mul_(inv)
            else:
                g.copy_(torch.from_numpy(self.pull_bufs[b] * inv))
        self.step_count += 1
