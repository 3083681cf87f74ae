"""Elastic rendezvous transitions (mirrors reference horovod_rendezvous_test.py)."""

from elasticdl_amd.master.rendezvous import ElasticRendezvousServer


def make_rdzv():
    r = ElasticRendezvousServer("127.0.0.1")
    r._flip_delay_sec = 0.0
    r._port = 1234  # don't start a real TCPStore for state-machine tests
    return r


def test_initial_world_forms_on_first_query():
    r = make_rdzv()
    r.add_worker("w0")
    r.add_worker("w1")
    info = r.get_comm_rank("w0")
    assert info["rendezvous_id"] == 1
    assert info["world_size"] == 2
    assert info["rank_id"] == 0
    assert r.get_comm_rank("w1")["rank_id"] == 1


def test_add_worker_flips_after_current_ready():
    r = make_rdzv()
    r.add_worker("w0")
    r.get_comm_rank("w0")  # world 1 = [w0]; w0 ready -> completed
    r.add_worker("w1")
    info = r.get_comm_rank("w1")
    assert info["rendezvous_id"] == 2
    assert info["world_size"] == 2
    assert info["rank_id"] == 1


def test_staged_removal():
    r = make_rdzv()
    r.add_worker("w0")
    r.add_worker("w1")
    assert r.get_comm_rank("w0")["rendezvous_id"] == 1
    assert r.get_comm_rank("w1")["rank_id"] == 1  # both ready -> completed
    r.remove_worker("w1")
    info = r.get_comm_rank("w0")
    assert info["rendezvous_id"] == 2
    assert info["world_size"] == 1
    assert r.get_comm_rank("w1")["rank_id"] == -1


def test_unknown_host_gets_minus_one():
    r = make_rdzv()
    r.add_worker("w0")
    r.get_comm_rank("w0")
    assert r.get_comm_rank("stranger")["rank_id"] == -1


def test_removal_before_ready_coalesces():
    r = make_rdzv()
    r.add_worker("w0")
    r.add_worker("w1")
    r.add_worker("w2")
    assert r.get_comm_rank("w0")["world_size"] == 3
    # w1 dies before the world completes: the dead member is exempt from
    # readiness, so once the remaining members have polled the staged
    # world [w0, w2] flips (gen 2)
    r.remove_worker("w1")
    info = r.get_comm_rank("w2")
    assert info["rendezvous_id"] == 2
    assert info["world_size"] == 2
    assert r.get_comm_rank("w0")["rank_id"] == 0


def test_world_resurrects_after_total_wipeout():
    """All workers die, a relaunched worker must be able to re-form the
    world (deviation from the reference, which refuses — its master would
    have killed the job; ours relaunches workers)."""
    r = make_rdzv()
    r.add_worker("w0")
    r.add_worker("w1")
    r.get_comm_rank("w0")
    r.get_comm_rank("w1")  # world 1 complete
    r.remove_worker("w0")
    r.remove_worker("w1")
    # any poll flips to the empty world
    assert r.get_comm_rank("w0")["rank_id"] == -1
    assert r.world_size() == 0
    # relaunched worker joins; empty world is trivially complete -> flip
    r.add_worker("w2")
    info = r.get_comm_rank("w2")
    assert info["rank_id"] == 0
    assert info["world_size"] == 1
    assert info["rendezvous_id"] >= 3


def test_relaunched_worker_joins_when_all_members_dead():
    """The exact flake scenario: every member of the current world died
    (removed) before reporting ready; a relaunched NON-member's poll must
    complete the world so the staged one flips."""
    r = make_rdzv()
    r.add_worker("w0")
    r.add_worker("w1")
    assert r.get_comm_rank("w0")["world_size"] == 2  # gen 1 formed
    # neither member ever polls again (both die); master stages removals
    r.remove_worker("w1")
    r.remove_worker("w0")
    r.add_worker("w2")  # relaunched worker
    info = r.get_comm_rank("w2")  # its poll alone must drive the flip(s)
    if info["rank_id"] < 0:  # at most one more poll needed
        info = r.get_comm_rank("w2")
    assert info["rank_id"] == 0
    assert info["world_size"] == 1


def test_real_tcpstore_start():
    r = ElasticRendezvousServer("127.0.0.1")
    port = r.start()
    assert port > 0
    # a client can connect and set/get through the store
    from datetime import timedelta

    from torch.distributed import TCPStore

    client = TCPStore("127.0.0.1", port, is_master=False, timeout=timedelta(seconds=10))
    client.set("k", "v")
    assert client.get("k") == b"v"
