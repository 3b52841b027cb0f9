"""Comm wrapper unit behavior without a process group (world=1 no-op
paths used by the self-PS mode and by every single-process entry)."""

import torch

from atomo_amd.parallel import Comm


def _comm():
    c = Comm(device=torch.device("cpu"))
    assert not c._initialized and c.world == 1
    return c


def test_noop_collectives_preserve_and_copy():
    c = _comm()
    t = torch.arange(4.0)
    c.broadcast(t)          # no-op
    c.reduce_sum(t)         # no-op
    c.all_reduce_sum(t)     # no-op
    assert torch.equal(t, torch.arange(4.0))
    out = torch.zeros(1, 4)
    c.gather(t, out)        # degenerates to a row copy
    assert torch.equal(out[0], t)


def test_noop_gather_partial_and_arrival():
    c = _comm()
    send = torch.full((6,), 3.0)
    bufs = torch.zeros(c.PIPE_DEPTH, 1, 6)
    seen = []
    n = c.gather_partial(send, bufs, step=0,
                         on_arrival=lambda w, s: seen.append((w, s)))
    assert n == 1 and seen == [(0, 0)]
    assert torch.equal(bufs[0, 0], send)
    stacked = torch.zeros(1, 6)
    order = c.gather_arrival(send, stacked)
    assert order == [0] and torch.equal(stacked[0], send)


def test_noop_partial_helpers_are_safe():
    c = _comm()
    c.publish_step(3)            # no-op without a group
    assert c.ps_step_behind(0) == 0
    c.drain_partial()            # nothing pending
    c.barrier()
    c.close()
