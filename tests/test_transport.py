"""Shm-ring transport engine tests."""

import glob
import os
import threading

import pytest

from fiber_amd.transport import ShmRing, Socket, new_address


def _spill_segments(ring_name):
    return glob.glob("/dev/shm/%s.sp.*" % ring_name)


class TestShmRing:
    def test_roundtrip(self):
        name = new_address("fam-t")
        ring = ShmRing(name, True, 1 << 20, 5.0)
        try:
            assert ring.send(b"abc", 1.0)
            assert ring.recv(1.0) == b"abc"
        finally:
            ring.close()
            ring.unlink()

    def test_wraparound(self):
        name = new_address("fam-t")
        ring = ShmRing(name, True, 64 << 10, 5.0)
        try:
            msg = b"m" * 5000
            for _ in range(100):
                ring.send(msg, 1.0)
                assert ring.recv(1.0) == msg
        finally:
            ring.close()
            ring.unlink()

    def test_blocking_backpressure(self):
        name = new_address("fam-t")
        ring = ShmRing(name, True, 16 << 10, 5.0)
        try:
            msg = b"x" * 4000
            sent = 0
            while ring.send(msg, 0.0):
                sent += 1
            assert 2 <= sent <= 4  # capacity-bounded
            assert ring.recv(1.0) == msg
            assert ring.send(msg, 1.0)  # space freed
        finally:
            ring.close()
            ring.unlink()

    def test_oversize_message_spills_and_roundtrips(self):
        """A message past ring capacity rides a spill segment: it still
        round-trips byte-exact and leaves no segment behind."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, 4 << 10, 5.0)
        try:
            for size in (8 << 10, 1 << 20, 10 << 20):
                msg = os.urandom(size)
                assert ring.send(msg, 5.0)
                assert ring.recv(5.0) == msg
            assert not _spill_segments(name)
        finally:
            ring.close()
            ring.unlink()

    def test_spill_mixed_with_inline_in_order(self):
        """FIFO order is preserved across inline records and spill
        control records, including through the batched ops."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, 8 << 10, 5.0)
        try:
            msgs = [b"s" * 100, os.urandom(64 << 10), b"t" * 200,
                    os.urandom(3 << 20), b"u" * 50]
            assert ring.send_many(msgs, 5.0) == len(msgs)
            got = list(ring.recv_many(3, 5.0))
            got += [ring.recv(5.0) for _ in range(len(msgs) - len(got))]
            assert got == msgs
            assert not _spill_segments(name)
        finally:
            ring.close()
            ring.unlink()

    def test_spill_peek_and_recv_into(self):
        """peek_size reports the REAL payload size of a spilled message;
        recv_into with a small buffer declines without consuming."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, 4 << 10, 5.0)
        try:
            msg = os.urandom(256 << 10)
            assert ring.send(msg, 5.0)
            assert ring.peek_size(1.0) == len(msg)
            small = bytearray(1024)
            n = ring.recv_into(small, 1.0)
            assert n == -(len(msg)) - 2  # too small, left in place
            big = bytearray(len(msg))
            assert ring.recv_into(big, 1.0) == len(msg)
            assert bytes(big) == msg
            assert not _spill_segments(name)
        finally:
            ring.close()
            ring.unlink()

    def test_spill_budget_backpressure(self):
        """Spilled payloads bypass ring-capacity backpressure, so the
        number of outstanding spill segments is capped (16): a producer
        of huge messages blocks instead of filling /dev/shm."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, 8 << 10, 5.0)
        try:
            big = os.urandom(64 << 10)
            sent = 0
            while ring.send(big, 0.0):
                sent += 1
            assert sent == 16, sent
            assert len(_spill_segments(name)) == 16
            assert ring.recv(1.0) == big  # frees one budget slot
            assert ring.send(big, 1.0)
            for _ in range(16):
                assert ring.recv(1.0) == big
            assert not _spill_segments(name)
        finally:
            ring.close()
            ring.unlink()

    def test_spill_sweep_on_unlink(self):
        """Unread spill segments are swept when the ring owner unlinks."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, 4 << 10, 5.0)
        ring.send(os.urandom(64 << 10), 5.0)
        ring.send(os.urandom(64 << 10), 5.0)
        assert len(_spill_segments(name)) == 2
        ring.close()
        ring.unlink()
        assert not _spill_segments(name)

    def test_unaligned_capacity_is_floored(self):
        """A capacity that is not a multiple of 8 is rounded down (the
        wrap marker needs 4 bytes before the end of the data area)."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, (4 << 10) + 5, 5.0)
        try:
            for _ in range(64):  # drift across the wrap point
                ring.send(b"z" * 500, 1.0)
                assert ring.recv(1.0) == b"z" * 500
        finally:
            ring.close()
            ring.unlink()

    def test_closed_ring_raises(self):
        name = new_address("fam-t")
        ring = ShmRing(name, True, 1 << 20, 5.0)
        ring.close()
        with pytest.raises(RuntimeError):
            ring.recv(1.0)
        ring.unlink()

    def test_open_missing_times_out(self):
        with pytest.raises(RuntimeError):
            ShmRing(new_address("fam-missing"), False, 0, 0.2)

    def test_mpmc_threads(self):
        name = new_address("fam-t")
        ring = ShmRing(name, True, 4 << 20, 5.0)
        n, nprod = 2000, 4
        received = []
        lock = threading.Lock()

        def produce(k):
            for i in range(n):
                ring.send(b"%d:%d" % (k, i), -1.0)

        def consume():
            for _ in range(n):
                msg = ring.recv(-1.0)
                with lock:
                    received.append(msg)

        threads = [
            threading.Thread(target=produce, args=(k,)) for k in range(nprod)
        ] + [threading.Thread(target=consume) for _ in range(nprod)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(30)
        assert len(received) == n * nprod
        assert len(set(received)) == n * nprod
        ring.close()
        ring.unlink()


class TestSocket:
    def test_push_pull(self):
        addr = new_address()
        w = Socket("w", addr, bind=True)
        r = Socket("r", addr, bind=False)
        w.send(b"task")
        assert r.recv(1.0) == b"task"
        w.close()
        r.close()

    def test_req_rep(self):
        addr = new_address()
        rep = Socket("rep", addr, bind=True)
        req = Socket("req", addr, bind=False)
        req.send(b"gimme")
        ident, payload = rep.recv_request(1.0)
        assert payload == b"gimme"
        assert ident == req.ident
        rep.send_reply(ident, b"task-1")
        assert req.recv(1.0) == b"task-1"
        req.close()
        rep.close()

    def test_rw_pair(self):
        addr = new_address()
        a = Socket("rw", addr, bind=True)
        b = Socket("rw", addr, bind=False)
        a.send(b"ping")
        assert b.recv(1.0) == b"ping"
        b.send(b"pong")
        assert a.recv(1.0) == b"pong"
        a.close()
        b.close()

    def test_bad_mode_raises(self):
        with pytest.raises(ValueError):
            Socket("xyz", "addr")


def _flood_producer(addr, payload_size, stop_after):
    from fiber_amd.transport import Socket

    sock = Socket("w", addr, bind=False)
    blob = b"p" * payload_size
    for _ in range(stop_after):
        sock.send(blob, timeout=5.0)


class TestDeadWriterReclaim:
    def test_consumer_survives_producer_sigkill_mid_stream(self):
        """A producer SIGKILLed while streaming large payloads must not
        wedge the ring: the reserve/commit protocol + dead-writer reclaim
        keeps other producers' messages flowing."""
        import time

        import fiber_amd
        from fiber_amd.transport import Socket, new_address

        addr = new_address("fam-chaos")
        reader = Socket("r", addr, bind=True)
        victim = fiber_amd.Process(
            target=_flood_producer, args=(addr, 100_000, 1_000_000)
        )
        survivor = fiber_amd.Process(
            target=_flood_producer, args=(addr, 100_000, 200)
        )
        victim.start()
        survivor.start()
        # let them stream, then SIGKILL the victim mid-flight
        for _ in range(20):
            assert reader.recv(timeout=10.0) is not None
        victim.kill()
        # drain: within a bounded time we must observe the survivor's
        # full 200 messages despite the victim's corpse in the ring
        got = 0
        deadline = time.monotonic() + 30.0
        while got < 100 and time.monotonic() < deadline:
            if reader.recv(timeout=5.0) is not None:
                got += 1
        assert got >= 100, "ring wedged after producer SIGKILL"
        victim.join(10)
        survivor.join(30)
        reader.close()


class TestBatchedOps:
    """send_many/recv_many: one lock hold + one wake per burst."""

    def test_batch_roundtrip_and_interleave(self):
        name = new_address("fam-t")
        ring = ShmRing(name, True, 1 << 20, 5.0)
        try:
            assert ring.send_many([b"a", b"bb", b"", b"dddd"], 1.0) == 4
            assert ring.recv(0.5) == b"a"
            assert ring.recv_many(10, 0.5) == [b"bb", b"", b"dddd"]
            ring.send(b"x1")
            ring.send_many([b"x2", b"x3"])
            assert ring.recv_many(2, 0.5) == [b"x1", b"x2"]
            assert ring.recv_many(5, 0.5) == [b"x3"]
            assert ring.recv_many(4, 0.0) == []
            assert ring.total_in == 7 and ring.total_out == 7
        finally:
            ring.close()
            ring.unlink()

    def test_batch_backpressure_order_exact(self):
        """A 600-message batch through a 4 KB ring: blocking bursts,
        wraps, exact content and counters."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, 4 << 10, 5.0)
        try:
            msgs = [bytes([i % 256]) * 100 for i in range(600)]
            sent = []
            th = threading.Thread(
                target=lambda: sent.append(ring.send_many(msgs, 30.0))
            )
            th.start()
            out = []
            while len(out) < 600:
                got = ring.recv_many(64, 5.0)
                assert got, "starved at %d" % len(out)
                out.extend(got)
            th.join(10)
            assert sent == [600]
            assert out == msgs
            assert ring.size == 0
        finally:
            ring.close()
            ring.unlink()

    def test_wrap_waste_near_full_geometry(self):
        """Regression: the producer fit check must price the wrap waste
        (capacity - tail) exactly.  Messages sized near capacity/3 keep
        the ring in the near-full wrap window where a fixed-margin check
        overruns the reader's head (used > capacity, unsigned underflow,
        corruption)."""
        import os
        import random

        name = new_address("fam-t")
        ring = ShmRing(name, True, 4 << 10, 5.0)
        try:
            random.seed(11)
            msgs = [os.urandom(random.randint(900, 1900)) for _ in range(300)]
            th = threading.Thread(target=lambda: ring.send_many(msgs, 60.0))
            th.start()
            out = []
            while len(out) < 300:
                got = ring.recv_many(8, 5.0)
                assert got, "starved at %d" % len(out)
                out.extend(got)
            th.join(10)
            assert out == msgs
            # singles through the same geometry
            th = threading.Thread(
                target=lambda: [ring.send(m, 60.0) for m in msgs]
            )
            th.start()
            out = [ring.recv(5.0) for _ in range(300)]
            th.join(10)
            assert out == msgs
        finally:
            ring.close()
            ring.unlink()

    def test_large_message_fits_regardless_of_tail_offset(self):
        """Empty-ring rewind: a near-capacity message must fit even after
        head/tail drifted to an arbitrary offset."""
        name = new_address("fam-t")
        ring = ShmRing(name, True, 4 << 10, 5.0)
        try:
            big = b"B" * ((4 << 10) - 64)
            for _ in range(8):  # drift the offsets, then send big
                ring.send(b"pad" * 41, 1.0)
                assert ring.recv(1.0)
                assert ring.send(big, 1.0), "big message did not fit"
                assert ring.recv(1.0) == big
        finally:
            ring.close()
            ring.unlink()

    def test_batch_survives_dead_writer_reclaim(self):
        """A producer SIGKILLed somewhere inside send_many must not stall
        batched consumers: reserved records from the dead pid are
        reclaimed, committed ones are delivered intact."""
        import os
        import signal

        name = new_address("fam-t")
        ring = ShmRing(name, True, 32 << 10, 5.0)
        try:
            pid = os.fork()
            if pid == 0:  # child: batch-send forever until killed
                child = ShmRing(name, False, 0, 5.0)
                batch = [b"k" * 64] * 16
                while True:
                    child.send_many(batch, 5.0)
            got = 0
            while got < 200:  # let plenty through first
                got += len(ring.recv_many(32, 5.0))
            os.kill(pid, signal.SIGKILL)
            os.waitpid(pid, 0)
            # live producer must still make progress afterwards
            assert ring.send_many([b"after1", b"after2"], 5.0) == 2
            seen = []
            deadline = 50
            while deadline and seen[-2:] != [b"after1", b"after2"]:
                seen.extend(ring.recv_many(32, 1.0))
                deadline -= 1
            assert seen[-2:] == [b"after1", b"after2"]
            for m in seen[:-2]:
                assert m == b"k" * 64
        finally:
            ring.close()
            ring.unlink()
