"""Model-based transport check: an ShmRing driven by random op sequences
must behave exactly like a FIFO deque (content, order, counters), across
single and batched ops, arbitrary payload sizes and wrap geometry."""

import collections
import os

from hypothesis import given, settings, strategies as st

from fiber_amd.transport import ShmRing, new_address

CAP = 4096  # small: forces wraps + backpressure in nearly every run

# max_size deliberately exceeds CAP: payloads past capacity/4 ride the
# spill path (one-shot shm segments), so the model covers inline records,
# wrap markers AND spill control records in the same FIFO.
payloads = st.binary(min_size=0, max_size=9000)
ops = st.lists(
    st.one_of(
        st.tuples(st.just("send"), payloads),
        st.tuples(st.just("send_many"),
                  st.lists(payloads, min_size=1, max_size=10)),
        st.tuples(st.just("recv"), st.none()),
        st.tuples(st.just("recv_many"),
                  st.integers(min_value=1, max_value=8)),
        st.tuples(st.just("peek"), st.none()),
    ),
    min_size=1,
    max_size=60,
)


class TestRingModel:
    @settings(max_examples=150, deadline=None)
    @given(ops=ops)
    def test_matches_fifo_model(self, ops):
        ring = ShmRing(new_address("fam-hyp"), True, CAP, 5.0)
        model = collections.deque()
        t_in = t_out = 0
        try:
            for op, arg in ops:
                if op == "send":
                    ok = ring.send(arg, 0.0)
                    # non-blocking: accepted iff it fit; mirror by result
                    if ok:
                        model.append(arg)
                        t_in += 1
                elif op == "send_many":
                    sent = ring.send_many(arg, 0.0)
                    assert sent <= len(arg)
                    for p in arg[:sent]:
                        model.append(p)
                    t_in += sent
                elif op == "recv":
                    got = ring.recv(0.0)
                    if model:
                        assert got == model.popleft()
                        t_out += 1
                    else:
                        assert got is None
                elif op == "recv_many":
                    got = ring.recv_many(arg, 0.0)
                    assert len(got) <= arg
                    if model:
                        assert got, "ring empty but model has %d" % len(model)
                    for g in got:
                        assert g == model.popleft()
                    t_out += len(got)
                elif op == "peek":
                    n = ring.peek_size(0.0)
                    if model:
                        assert n == len(model[0])
                    else:
                        assert n == -1
                assert ring.size == len(model)
                assert ring.total_in == t_in
                assert ring.total_out == t_out
            # drain and verify the remainder
            while model:
                assert ring.recv(0.0) == model.popleft()
        finally:
            ring.close()
            ring.unlink()
