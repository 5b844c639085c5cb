"""The reference's acceptance suite, MPI-free.

Same three tests and assertions as /root/reference/test/test_cgx.py:69-101
(constant-tensor exactness across sizes/dtypes/bits including the int32
passthrough, the analytic L-inf error bound for ramp inputs, and bits=32
exactness), driven through `dist.all_reduce` with env-var reconfiguration
between sweeps — launched with torchrun instead of mpirun:

    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
        -m pytest test/test_cgx.py -q
"""

import os

os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
import unittest

import numpy as np
import torch
import torch.distributed as dist

import torch_cgx_amd  # noqa: F401  (registers "cgx")


class CGXTests(unittest.TestCase):
    @classmethod
    def setUpClass(cls):
        if "WORLD_SIZE" not in os.environ:
            raise unittest.SkipTest("launch with torchrun (see module doc)")
        if not torch.cuda.is_available():
            raise unittest.SkipTest("needs GPUs")
        cls.rank = int(os.environ["RANK"])
        cls.world_size = int(os.environ["WORLD_SIZE"])
        dist.init_process_group("cgx", init_method="env://",
                                rank=cls.rank, world_size=cls.world_size)
        torch.cuda.set_device(cls.rank % torch.cuda.device_count())

    @classmethod
    def tearDownClass(cls):
        dist.barrier()
        dist.destroy_process_group()

    def test_compressed_exact(self):
        ws = self.world_size
        for q in [2, 4, 8]:
            os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = str(q)
            for dtype in [torch.float16, torch.float32, torch.int32]:
                for size in [1, 2, 8, 128, 1024, 1000000]:
                    expected = torch.tensor(
                        [(ws * (ws + 1)) // 2] * size, dtype=dtype,
                        device="cuda")
                    for _ in range(10):
                        t = torch.tensor([self.rank + 1] * size, dtype=dtype,
                                         device="cuda")
                        dist.all_reduce(t)
                        assert torch.equal(t, expected), (q, dtype, size)

    def test_compressed_non_exact(self):
        ws = self.world_size
        for q in [2, 3, 4, 6, 8]:
            os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = str(q)
            for dtype in [torch.float16, torch.float32]:
                for size in [128, 1024, 1025, 16384, 1000000]:
                    arange = np.arange(-size / 2, size / 2, 1.0)
                    if dtype == torch.float16:
                        arange = arange * 1e-3
                    expected = torch.tensor(
                        (ws * (ws + 1) / 2) * arange, dtype=dtype,
                        device="cuda")
                    for bucket_size in [64, 512, 2048]:
                        os.environ["CGX_COMPRESSION_BUCKET_SIZE"] = \
                            str(bucket_size)
                        for _ in range(10):
                            t = torch.tensor((self.rank + 1) * arange,
                                             dtype=dtype, device="cuda")
                            dist.all_reduce(t)
                            coef = ws * (ws + 1)
                            bound = (2 * min(bucket_size, size)
                                     / ((1 << q) - 1) * coef)
                            err = torch.norm(t - expected,
                                             p=float("inf")).item()
                            assert err < bound, (q, bucket_size, size, err)

    def test_uncompressed(self):
        ws = self.world_size
        os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = "32"
        for dtype in [torch.float16, torch.float32, torch.int32]:
            for size in [1, 2, 8, 128, 1024, 1000000]:
                t = torch.tensor([self.rank + 1] * size, dtype=dtype,
                                 device="cuda")
                dist.all_reduce(t)
                expected = torch.tensor([(ws * (ws + 1)) // 2] * size,
                                        dtype=dtype, device="cuda")
                assert torch.equal(t, expected)


if __name__ == "__main__":
    unittest.main()
