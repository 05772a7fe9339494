"""Multi-host cluster bring-up: the Rabit tracker replacement.

The reference forms its cluster with a vendored DMLC tracker (TCP rank
brokering, tree/ring maps — dmlc_patch/tracker.py) plus
`collective.CommunicatorContext` retries (distributed.py:172-234). On
MI355X none of that machinery is needed: torch.distributed's TCP rendezvous
bootstraps the communicator directly — gloo across hosts for the control
plane and CPU fallback, RCCL (backend "nccl") over xGMI for in-node GPU
ranks. This module keeps the reference's *semantics*:

  * wait_hostname_resolution — retrying DNS for all hosts (15 min cap);
  * two-phase `rabit_run` (distributed.py:42-109): form the full cluster,
    broadcast which hosts actually hold data, then re-form the cluster with
    only the data-holding hosts and run `exec_fun` there (other hosts exit
    cleanly);
  * `RabitHelper.synchronize` — every rank shares a JSON-able payload.

API names (rabit_run, Rabit, wait_hostname_resolution) are kept so script
users migrating from the reference find the same surface.
"""
import datetime
import logging
import socket
import sys
import time

import torch.distributed as dist

from ..toolkit import exceptions as exc

logger = logging.getLogger(__name__)

LOCAL_HOSTNAME = "127.0.0.1"
DEFAULT_PORT = 9099


def _dns_lookup(host):
    return socket.gethostbyname(host)


def wait_hostname_resolution(sm_hosts, max_wait_s=900):
    """Block until every host in the cluster resolves (reference :30-39)."""
    deadline = time.time() + max_wait_s
    for host in sm_hosts:
        while True:
            try:
                _dns_lookup(host)
                break
            except socket.gaierror:
                if time.time() > deadline:
                    raise exc.PlatformError(f"Could not resolve hostname {host} within {max_wait_s}s")
                time.sleep(1)


class RabitHelper:
    """Info/utility object handed to the training function."""

    def __init__(self, is_master, current_host, master_port, comm=None):
        self.is_master = is_master
        self.rank = comm.rank if comm else 0
        self.current_host = current_host
        self.master_port = master_port
        self.comm = comm

    def synchronize(self, data):
        """Collect `data` from every rank; returns the list of results."""
        if self.comm is None:
            return [data]
        return self.comm.allgather_object(data)


class Rabit:
    """Context manager forming one torch.distributed cluster over TCP.

    Master is hosts[0] (reference semantics: distributed.py:150-170). On a
    single host this degenerates to a no-op cluster.
    """

    def __init__(self, hosts, current_host=None, master_host=None, port=None, backend="gloo",
                 timeout_s=1800):
        self.hosts = sorted(hosts)
        self.n_workers = len(self.hosts)
        self.port = port or DEFAULT_PORT
        self.current_host = current_host or socket.gethostname()
        self.master_host = master_host or self.hosts[0]
        self.backend = backend
        self.timeout_s = timeout_s
        self.is_master_host = self.current_host == self.master_host
        self.rank = self.hosts.index(self.current_host)
        self.comm = None

    def start(self):
        from . import comm as comm_mod

        if self.n_workers == 1:
            logger.debug("Single host cluster; no process group needed")
            return RabitHelper(True, self.current_host, self.port, None)
        master_ip = _dns_lookup(self.master_host)
        logger.info(
            "Connecting to cluster master %s:%s as rank %d/%d",
            master_ip, self.port, self.rank, self.n_workers,
        )
        # startup robustness: retry the rendezvous (reference retries
        # CommunicatorContext init, distributed.py:215-227)
        last_error = None
        for attempt in range(3):
            try:
                dist.init_process_group(
                    backend=self.backend,
                    init_method=f"tcp://{master_ip}:{self.port}",
                    rank=self.rank,
                    world_size=self.n_workers,
                    timeout=datetime.timedelta(seconds=self.timeout_s),
                )
                last_error = None
                break
            except Exception as e:  # noqa: BLE001 - connection errors vary by backend
                last_error = e
                logger.warning("Cluster rendezvous attempt %d failed: %s", attempt + 1, e)
                time.sleep(2 * (attempt + 1))
        if last_error is not None:
            raise exc.PlatformError("Could not join the training cluster", caused_by=last_error)
        self.comm = comm_mod.Communicator()
        return RabitHelper(self.rank == 0, self.current_host, self.port, self.comm)

    def stop(self):
        if self.comm is not None and dist.is_initialized():
            dist.barrier()
            dist.destroy_process_group()
            self.comm = None

    def __enter__(self):
        return self.start()

    def __exit__(self, exc_type, exc_value, exc_traceback):
        self.stop()


def rabit_run(
    exec_fun,
    args,
    include_in_training,
    hosts,
    current_host,
    first_port=None,
    second_port=None,
    update_rabit_args=False,
):
    """Two-phase cluster formation (reference distributed.py:42-109).

    Phase 1: all hosts join; each broadcasts whether it has data.
    Phase 2: only data-holding hosts re-form the cluster and train; hosts
    without data exit 0.
    """
    first_port = first_port or DEFAULT_PORT
    second_port = second_port or first_port + 1

    with Rabit(hosts=hosts, current_host=current_host, port=first_port) as helper:
        hosts_with_data = [
            record["host"]
            for record in helper.synchronize({"host": helper.current_host, "include_in_training": include_in_training})
            if record["include_in_training"]
        ]
        hosts_with_data.sort()
        if not hosts_with_data:
            raise exc.UserError("No hosts have training data: cannot run distributed training")

    if not include_in_training:
        logger.warning("Host %s not being used for distributed training.", current_host)
        sys.exit(0)

    if len(hosts_with_data) == len(hosts) and len(hosts) == 1:
        # single host fast path
        if update_rabit_args:
            args = dict(args, is_master=True)
        return exec_fun(**args)

    with Rabit(hosts=hosts_with_data, current_host=current_host, port=second_port) as helper:
        if update_rabit_args:
            args = dict(args, is_master=helper.is_master, comm=helper.comm)
        return exec_fun(**args)
