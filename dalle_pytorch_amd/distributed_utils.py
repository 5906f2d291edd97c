"""Reference-compatible distributed facade.

The reference exposes a pluggable backend registry
(dalle_pytorch/distributed_utils.py:22-96) with DeepSpeed / Horovod / Dummy
backends. This framework replaces the zoo with one RCCL engine
(`dalle_pytorch_amd.parallel`); this module keeps the reference's module
surface (`wrap_arg_parser`, `set_backend_from_args`, `using_backend`,
`backend`, `is_distributed`) so code written against the reference's
distributed layer keeps working — every choice maps onto the RCCL engine.
"""

import torch

from dalle_pytorch_amd import parallel


class RCCLBackend:
    """The reference DistributedBackend contract
    (distributed_backends/distributed_backend.py:12-178) on the RCCL engine."""

    BACKEND_MODULE_NAME = 'torch.distributed'
    BACKEND_NAME = 'RCCL'

    def __init__(self):
        self._initialized = False

    # lifecycle -----------------------------------------------------------
    def initialize(self):
        parallel.init_distributed()
        self._initialized = True

    def is_initialized(self):
        return self._initialized

    # topology ------------------------------------------------------------
    def get_world_size(self):
        return parallel.get_world_size()

    def get_rank(self):
        return parallel.get_rank()

    def get_local_rank(self):
        return parallel.get_local_rank()

    def is_root_worker(self):
        return parallel.get_rank() == 0

    def is_local_root_worker(self):
        return parallel.get_local_rank() == 0

    def local_barrier(self):
        parallel.barrier()

    def check_batch_size(self, batch_size):
        assert batch_size >= self.get_world_size(), \
            (f'batch size can not be smaller than number of processes '
             f'({batch_size} < {self.get_world_size()})')

    # wrapping ------------------------------------------------------------
    def distribute(self, _args=None, model=None, optimizer=None,
                   model_parameters=None, training_data=None,
                   lr_scheduler=None, **_kwargs):
        engine = parallel.DataParallelEngine(model)
        model._dp_engine = engine  # kept reachable for finish_gradient_sync
        return model, optimizer, training_data, lr_scheduler

    def average_all(self, tensor):
        return parallel.average_scalar(tensor)


class DummyBackend(RCCLBackend):
    """Single-process stand-in (reference dummy_backend.py)."""

    BACKEND_NAME = 'Dummy'

    def initialize(self):
        self._initialized = True

    def get_world_size(self):
        return 1

    def get_rank(self):
        return 0

    def get_local_rank(self):
        return 0

    def local_barrier(self):
        pass

    def distribute(self, _args=None, model=None, optimizer=None,
                   model_parameters=None, training_data=None,
                   lr_scheduler=None, **_kwargs):
        return model, optimizer, training_data, lr_scheduler

    def average_all(self, tensor):
        return tensor


# aliases so reference-style `using_backend(DeepSpeedBackend)` checks resolve
DeepSpeedBackend = RCCLBackend
HorovodBackend = RCCLBackend

BACKENDS = [RCCLBackend, DummyBackend]

backend = None
is_distributed = None


def wrap_arg_parser(parser):
    """Reference CLI surface (distributed_utils.py:34-45): --distributed_backend
    plus the legacy --deepspeed flag; both select the RCCL engine."""
    parser.add_argument('--deepspeed', action='store_true', default=None,
                        help='legacy alias: use the distributed engine')
    parser.add_argument('--distributed_backend', '--distr_backend',
                        type=str, default=None,
                        help="any of 'rccl'/'deepspeed'/'horovod' -> the RCCL "
                             "engine; None -> single process")
    return parser


def set_backend_from_args(args):
    global backend, is_distributed
    name = getattr(args, 'distributed_backend', None)
    if name is None and getattr(args, 'deepspeed', None):
        name = 'deepspeed'
    if name is None or int(torch.cuda.is_available()) == 0 and name == '':
        backend = DummyBackend()
        is_distributed = False
    else:
        backend = RCCLBackend()
        is_distributed = True
    return backend


def require_set_backend():
    assert backend is not None, \
        'distributed backend is not set; call set_backend_from_args first'


def using_backend(test_backend):
    require_set_backend()
    if isinstance(test_backend, str):
        return backend.BACKEND_NAME.lower() == test_backend.lower()
    return isinstance(backend, test_backend)
