"""NASNet conv-path step time: native SepConv (depthwise kernel +
batched-GEMM pointwise) vs library ops, channels aligned (filters=32)."""
import sys
import time

import torch

sys.path.insert(0, ".")


def main():
    from adanet_amd.models.nasnet import NasNetCIFAR
    from adanet_amd.ops.linear import restore_fp32_params
    from adanet_amd.ops.optim import FusedSGD
    dev = "cuda:0"
    torch.manual_seed(0)
    for filters in (32, 10):
        m = NasNetCIFAR(num_cells=3, num_conv_filters=filters).to(dev)
        m.to(torch.bfloat16)
        restore_fp32_params(m)
        opt = FusedSGD(m.parameters(), lr=0.01, momentum=0.9)
        x = torch.randn(96, 3, 32, 32, device=dev).to(torch.bfloat16)
        yl = torch.randint(0, 10, (96,), device=dev)

        def step():
            opt.zero_grad(set_to_none=True)
            last, logits = m(x)
            loss = torch.nn.functional.cross_entropy(logits.float(), yl)
            loss.backward()
            opt.step()

        for _ in range(5):
            step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            step()
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / 10 * 1e3
        mode = ("fully-native" if filters % 32 == 0 else
                "native-depthwise (GEMM paths unaligned -> library)")
        import os
        if os.environ.get("ADANET_NATIVE_CONV") == "0":
            mode = "all-library (MIOpen)"
        print("filters=%d [%s]: %.1f ms/step" % (filters, mode, ms))


if __name__ == "__main__":
    main()
