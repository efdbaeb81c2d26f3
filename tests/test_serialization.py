"""Serialization policy tests (host side; the HIP-IPC tensor path is
covered by tests/gpu/test_ipc.py)."""

import pytest
import torch

from fiber_amd import serialization


def module_level_fn(x):
    return x + 1


class TestSerialization:
    def test_roundtrip_plain(self):
        for obj in [1, "s", [1, 2], {"k": (3, 4)}, None, b"bytes"]:
            assert serialization.loads(serialization.dumps(obj)) == obj

    def test_cpu_tensor_by_value(self):
        t = torch.arange(10, dtype=torch.float32)
        t2 = serialization.loads(serialization.dumps(t))
        assert torch.equal(t, t2)
        t2 += 1  # independent copy
        assert not torch.equal(t, t2)

    def test_function_by_reference(self):
        blob = serialization.dumps_closure(module_level_fn)
        fn = serialization.loads(blob)
        assert fn(2) == 3

    def test_closure_via_cloudpickle(self):
        y = 10

        def closure(x):
            return x + y

        blob = serialization.dumps_closure(closure)
        assert serialization.loads(blob)(5) == 15

    def test_lambda_via_cloudpickle(self):
        blob = serialization.dumps_closure(lambda x: x * 3)
        assert serialization.loads(blob)(4) == 12

    def test_loads_accepts_memoryview(self):
        blob = serialization.dumps({"a": 1})
        assert serialization.loads(memoryview(blob)) == {"a": 1}


class TestTransportRecvInto:
    def test_peek_and_recv_into(self):
        from fiber_amd.transport import ShmRing, new_address

        ring = ShmRing(new_address("fam-t"), True, 1 << 20, 5.0)
        try:
            ring.send(b"hello world", 1.0)
            assert ring.peek_size(1.0) == 11
            buf = bytearray(64)
            n = ring.recv_into(buf, 1.0)
            assert n == 11
            assert bytes(buf[:11]) == b"hello world"
        finally:
            ring.close()
            ring.unlink()

    def test_recv_into_grows(self):
        from fiber_amd.transport import ShmRing, new_address, ring_recv_view

        ring = ShmRing(new_address("fam-t"), True, 1 << 20, 5.0)
        try:
            payload = b"z" * (200 << 10)  # larger than the 64 KiB buffer
            ring.send(payload, 1.0)
            view = ring_recv_view(ring, 1.0)
            assert bytes(view) == payload
        finally:
            ring.close()
            ring.unlink()

    def test_recv_into_timeout(self):
        from fiber_amd.transport import ShmRing, new_address

        ring = ShmRing(new_address("fam-t"), True, 1 << 20, 5.0)
        try:
            buf = bytearray(16)
            assert ring.recv_into(buf, 0.0) == -1
            assert ring.peek_size(0.0) == -1
        finally:
            ring.close()
            ring.unlink()
