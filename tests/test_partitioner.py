from torchsnapshot_amd.partitioner import PartitionItem, _greedy_assign


def test_greedy_balances_loads():
    items = [PartitionItem(f"replicated/t{i}", 100) for i in range(8)]
    assignment = _greedy_assign(items, [0, 0, 0, 0])
    per_rank = [0, 0, 0, 0]
    for path, rank in assignment.items():
        per_rank[rank] += 100
    assert per_rank == [200, 200, 200, 200]


def test_greedy_respects_preloads():
    # rank 0 already has 1000 bytes of non-replicated writes
    items = [PartitionItem(f"p{i}", 100) for i in range(4)]
    assignment = _greedy_assign(items, [1000, 0])
    assert all(rank == 1 for rank in assignment.values())


def test_greedy_big_items_spread():
    items = [
        PartitionItem("big0", 1000),
        PartitionItem("big1", 1000),
        PartitionItem("small0", 10),
        PartitionItem("small1", 10),
    ]
    assignment = _greedy_assign(items, [0, 0])
    assert assignment["big0"] != assignment["big1"]


def test_allowed_ranks_constraint():
    items = [PartitionItem(f"p{i}", 100, allowed_ranks=[2, 3]) for i in range(4)]
    assignment = _greedy_assign(items, [0, 0, 0, 0])
    assert set(assignment.values()) == {2, 3}
