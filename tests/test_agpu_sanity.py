"""First-to-run GPU sanity (alphabetically before the kernel suites): if
this fails, the box/runtime/extension is broken — not a specific kernel.
Run order matters because the driver uses -x."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gpu_sanity_and_extension_load():
    assert torch.cuda.is_available(), "no GPU visible"
    props = torch.cuda.get_device_properties(0)
    print(f"device: {props.name}, {props.total_memory / 2**30:.0f} GiB, "
          f"{props.multi_processor_count} CUs")
    # plain torch op round-trips
    x = torch.arange(8, device="cuda:0", dtype=torch.float32)
    assert float((x * 2).sum().cpu()) == 56.0

    from dgl_operator_amd.ops import backend

    ext = backend.load_extension(required=True)
    assert ext is not None
    # smallest possible kernel exercise: gather 2 rows
    feat = torch.eye(4, device="cuda:0")
    rows = torch.tensor([2, 0], device="cuda:0")
    out = ext.gather_rows(feat, rows, None, 0)
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), feat.cpu()[[2, 0]])
