"""2-D device mesh (dp, tp) — counterpart of the reference's
init_device_mesh((num_nodes, gpus_on_node), ("dp", "tp"))
(/root/reference/06-tensor-parallel/train_llm.py:51-55,
07-2d-parallel/train_llm.py:47-53).  tp is the INNER (consecutive-rank)
dimension so tensor-parallel collectives stay on intra-node xGMI."""
import torch.distributed as dist


class DeviceMesh2D:
    def __init__(self, tp_size: int, world_size: int | None = None):
        world = world_size or dist.get_world_size()
        if world % tp_size != 0:
            raise ValueError(f"world {world} not divisible by tp {tp_size}")
        self.world_size = world
        self.tp_size = tp_size
        self.dp_size = world // tp_size
        rank = dist.get_rank()
        self.rank = rank
        self.dp_rank = rank // tp_size
        self.tp_rank = rank % tp_size

        # every rank must participate in every new_group call
        self.tp_group = None
        self.dp_group = None
        if tp_size == world and self.dp_size == 1:
            self.tp_group = dist.group.WORLD
        else:
            for d in range(self.dp_size):
                ranks = list(range(d * tp_size, (d + 1) * tp_size))
                g = dist.new_group(ranks)
                if rank in ranks:
                    self.tp_group = g
        if self.dp_size == world and tp_size == 1:
            self.dp_group = dist.group.WORLD
        else:
            for t in range(tp_size):
                ranks = list(range(t, world, tp_size))
                g = dist.new_group(ranks)
                if rank in ranks:
                    self.dp_group = g

    def __repr__(self):
        return (f"DeviceMesh2D(dp={self.dp_size}, tp={self.tp_size}, "
                f"rank={self.rank}->({self.dp_rank},{self.tp_rank}))")
