"""Cluster/launch utilities: resource_info parsing, logging, device discovery.

MI355X-native equivalent of the reference's common/lib.py (358 loc).  The
``resource_info`` FILE FORMAT is preserved exactly
(``host[:gpu,gpu,...]`` per line, first host = master — reference
lib.py:121-150, doc/quick_start.md), as is the ``^``/``+``/``:`` serialized
codec (lib.py:153-176).  What changes is discovery: instead of ssh-probing
``/proc/driver/nvidia/gpus`` the local machine is probed through the HIP
runtime (torch.cuda on ROCm / rocm-smi), and ports are reserved locally —
there is no ssh plane on a single MI355X node.
"""

import logging
import os
import socket
import sys

from parallax_amd import consts


# --- logging (reference lib.py:58-67) --------------------------------------
parallax_log = logging.getLogger('PARALLAX')
try:
    parallax_log.setLevel(int(os.environ["PARALLAX_LOG_LEVEL"]))
except (KeyError, ValueError):
    parallax_log.setLevel(logging.INFO)
_handler = logging.StreamHandler()
_handler.setFormatter(
    logging.Formatter('%(levelname)s:%(thread)d:%(name)s:%(message)s', None))
parallax_log.addHandler(_handler)
parallax_log.propagate = False


def local_gpu_count():
    """Number of visible HIP devices on this machine (0 on a CPU-only box)."""
    try:
        import torch
        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:  # pragma: no cover - torch import failure
        pass
    return 0


def _get_available_gpus(hostname):
    """Local-machine GPU probe (reference lib.py:101-103 ssh-probed nvidia;
    here: HIP device count)."""
    return list(range(local_gpu_count()))


def get_free_port():
    """Reserve an ephemeral TCP port on the local machine
    (reference lib.py:106-118 did this over ssh)."""
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


# --- resource_info file format (reference lib.py:121-150) -------------------
def parse_machine_info(machine_str):
    machine_str = machine_str.strip()
    if not machine_str or machine_str.startswith('#'):
        return []
    hostname_gpus = machine_str.split(':')
    hostname = hostname_gpus[0]
    if len(hostname_gpus) == 1:
        gpus = _get_available_gpus(hostname)
    else:
        gpus = [int(g) for g in hostname_gpus[1].split(',') if g != '']
    return [(hostname, gpus)]


def parse_resource_info(path_or_str, run_option='HYBRID'):
    """Parse a resource_info file (or a literal resource string containing
    newlines / no file at that path) into the reference's dict shape:
    {'master': [...], 'ps': [...], 'worker': [...]}, each entry
    {'hostname', 'port': [ports], 'gpus': [ids]}."""
    machines = []
    if os.path.isfile(path_or_str):
        with open(path_or_str) as f:
            lines = f.readlines()
    else:
        lines = str(path_or_str).splitlines()
    for machine_info in lines:
        machines.extend(parse_machine_info(machine_info))
    if not machines:
        raise ValueError('resource_info is empty: %r' % (path_or_str,))

    master_host = machines[0][0]
    master = [{'hostname': master_host, 'port': [get_free_port()],
               'gpus': []}]
    ps = [{'hostname': hostname, 'port': [get_free_port()], 'gpus': []}
          for hostname, _ in machines]
    worker = [{'hostname': hostname,
               'port': [get_free_port() for _ in
                        range(1 if run_option != 'HYBRID' or len(gpus) == 0
                              else len(gpus))],
               'gpus': gpus}
              for hostname, gpus in machines]
    return {'master': master, 'ps': ps, 'worker': worker}


# --- serialized codec (reference lib.py:153-176) ----------------------------
def serialize_resource_info(resource_info):
    def serialize_machine(m):
        return '%s:%s:%s' % (
            m['hostname'],
            ','.join(str(p) for p in m['port']),
            ','.join(str(g) for g in m['gpus']))

    def serialize_machines(machines):
        return '+'.join(serialize_machine(m) for m in machines)

    return '^'.join('%s_%s' % (typ, serialize_machines(machines))
                    for typ, machines in resource_info.items())


def deserialize_resource_info(resource_info_serialized):
    def deserialize_list(lst):
        if len(lst) == 0:
            return []
        return [int(g) for g in lst.split(',')]

    def deserialize_machine(m):
        hostname, ports, gpus = m.split(':')
        return {'hostname': hostname, 'port': deserialize_list(ports),
                'gpus': deserialize_list(gpus)}

    resource_info = {}
    for type_machine in resource_info_serialized.strip().split('^'):
        typ, machines = type_machine.split('_', 1)
        resource_info[typ] = [deserialize_machine(m)
                              for m in machines.split('+')]
    return resource_info


def num_total_workers(resource_info):
    """One worker per GPU (one process per device — the MI355X execution
    model); a GPU-less machine still contributes one CPU worker
    (reference runner.py:93-95 num_workers arithmetic)."""
    return sum(max(1, len(w['gpus'])) for w in resource_info['worker'])


# --- profile dir scaffolding (reference lib.py:333-358) ---------------------
def create_profile_directory(profile_dir, resource_info):
    os.makedirs(profile_dir, exist_ok=True)
    manifest = os.path.join(profile_dir, 'task_info')
    with open(manifest, 'w') as f:
        for i, w in enumerate(resource_info['worker']):
            f.write('worker:%d:%s:%s\n'
                    % (i, w['hostname'], ','.join(map(str, w['gpus']))))
    for i, w in enumerate(resource_info['worker']):
        os.makedirs(os.path.join(profile_dir, w['hostname'],
                                 'worker:%d' % i, 'run_meta'),
                    exist_ok=True)


def redirect_stdout_stderr(redirect_path, worker_id):
    """Per-process log capture (reference redirect_path,
    ps/runner.py:34-46)."""
    os.makedirs(redirect_path, exist_ok=True)
    out = open(os.path.join(redirect_path, 'worker_%d.out' % worker_id), 'a')
    err = open(os.path.join(redirect_path, 'worker_%d.err' % worker_id), 'a')
    os.dup2(out.fileno(), sys.stdout.fileno())
    os.dup2(err.fileno(), sys.stderr.fileno())
