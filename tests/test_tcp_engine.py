"""TCP weight-transfer engine round trip (reference TCPTransferEngine
capability): N parallel localhost streams, recv_into a
registered buffer, bitwise equality + async status surface."""
import torch

from polyrl_amd.transfer.tcp_engine import TcpWeightReceiver, TcpWeightSender


def test_roundtrip_bitwise():
    torch.manual_seed(0)
    nbytes = 3 * (1 << 20) + 137          # odd size exercises span split
    src = torch.randint(0, 256, (nbytes,), dtype=torch.uint8)
    dst = torch.zeros(nbytes, dtype=torch.uint8)

    rx = TcpWeightReceiver(dst, num_streams=4)
    rx.expect(nbytes)
    tx = TcpWeightSender(num_streams=4)
    bid = tx.submit(src, "127.0.0.1", rx.ports)
    assert tx.wait(bid, timeout=60.0) == "done"
    assert rx.wait(timeout=60.0)
    rx.close()
    assert torch.equal(src, dst)


def test_weight_state_dict_roundtrip():
    """Pack a state dict into a flat buffer, ship it, reconstruct views —
    the sender-agent / receiver-agent data path (fsdp_interface.py:186-207
    pack + patches.py:205-215 view reconstruction capability)."""
    torch.manual_seed(1)
    sd = {"a.weight": torch.randn(33, 17), "b.bias": torch.randn(129)}
    metas = [(k, v.shape, v.dtype) for k, v in sd.items()]
    flat = torch.cat([v.reshape(-1).view(torch.uint8).view(-1)
                      if v.dtype == torch.uint8 else
                      v.reshape(-1).float().view(torch.uint8).reshape(-1)
                      for v in sd.values()])
    dst = torch.zeros_like(flat)
    rx = TcpWeightReceiver(dst, num_streams=2)
    rx.expect(flat.numel())
    tx = TcpWeightSender(num_streams=2)
    bid = tx.submit(flat, "127.0.0.1", rx.ports)
    assert tx.wait(bid) == "done" and rx.wait()
    rx.close()
    off = 0
    out = {}
    for name, shape, dtype in metas:
        n = int(torch.tensor([], dtype=dtype).element_size()
                * torch.Size(shape).numel())
        out[name] = dst[off:off + n].view(torch.float32).view(shape)
        off += n
    for k in sd:
        assert torch.equal(out[k], sd[k].float())


def test_fp8_quantization_error_bound():
    """Per-tensor e4m3 + scale: worst-case relative error for normal
    weights stays under e4m3's ~6% mantissa step (the compress="fp8"
    payload math, transfer/tcp_engine.py)."""
    import torch
    torch.manual_seed(0)
    for shape in [(64, 64), (3, 1000), (17,)]:
        t = torch.randn(*shape) * 0.02          # weight-scale magnitudes
        scale = float(t.abs().amax().clamp(min=1e-12)) / 448.0
        q = (t.float() / scale).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
        back = q.to(torch.float32) * scale
        # elements above 1% of amax: bounded relative error
        big = t.abs() > 0.01 * t.abs().amax()
        rel = ((back - t).abs()[big] / t.abs()[big]).max().item()
        assert rel < 0.07, rel
        # bytes really halve vs bf16
        assert q.element_size() == 1


def test_oversized_span_fails_fast():
    """a sender span beyond the registered buffer must
    surface a transfer error from wait() immediately instead of hanging
    until the install timeout."""
    import socket
    import struct

    import pytest
    import torch

    from polyrl_amd.transfer.tcp_engine import TcpWeightReceiver

    buf = torch.zeros(1024, dtype=torch.uint8)
    rx = TcpWeightReceiver(buf, num_streams=1)
    rx.expect(1024)
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.connect(("127.0.0.1", rx.ports[0]))
    # header claims a span past the end of the buffer
    s.sendall(struct.Struct("<QQ").pack(512, 4096))
    with pytest.raises(RuntimeError, match="exceeds buffer"):
        rx.wait(timeout=10.0)
    s.close()
    rx.close()
