"""Multi-node runners: build the cross-node launch command for each
cluster flavor.

Parity: reference `deepspeed/launcher/multinode_runner.py` (PDSH:55,
OpenMPI:126, MPICH:188, IMPI:260, Slurm:345, MVAPICH:393). Runners only
CONSTRUCT commands (testable without a cluster); `runner.main` executes
them. MPI-family runners launch the user script directly under mpirun
(one process per slot, rank env derived from the MPI env by
comm.init_distributed's mpi discovery); pdsh/ssh/slurm fan out
deepspeed_amd.launcher.launch per node.
"""
import os
import shlex
import shutil
import sys


class MultiNodeRunner:
    name = "base"

    def __init__(self, args, world_info):
        self.args = args
        self.world_info = world_info  # {host: [gpu ids]}
        self.exports = {}

    def backend_exists(self):
        raise NotImplementedError

    def get_cmd(self, environment=None):
        raise NotImplementedError

    def add_export(self, key, var):
        self.exports[key.strip()] = str(var).strip()

    @property
    def user_arguments(self):
        return [self.args.user_script] + list(self.args.user_args)

    @property
    def nnodes(self):
        return len(self.world_info)

    @property
    def nprocs(self):
        return sum(len(g) for g in self.world_info.values())


class PDSHRunner(MultiNodeRunner):
    name = "pdsh"

    def backend_exists(self):
        return shutil.which("pdsh") is not None

    def get_cmd(self, environment=None):
        import json
        hosts = ",".join(self.world_info.keys())
        master = self.args.master_addr or next(iter(self.world_info))
        winfo = json.dumps(self.world_info)
        envs = "".join(f"export {k}={shlex.quote(v)}; "
                       for k, v in self.exports.items())
        node_cmd = (
            f"{envs}cd {os.getcwd()} && {sys.executable} -m "
            f"deepspeed_amd.launcher.launch "
            f"--master_addr={master} --master_port={self.args.master_port} "
            f"--world_info={shlex.quote(winfo)} "
            f"{self.args.user_script} "
            + " ".join(map(shlex.quote, self.args.user_args)))
        return ["pdsh", "-S", "-f", "1024", "-w", hosts, node_cmd]


class OpenMPIRunner(MultiNodeRunner):
    name = "openmpi"

    def backend_exists(self):
        return shutil.which("ompi_info") is not None

    def get_cmd(self, environment=None):
        cmd = ["mpirun", "-n", str(self.nprocs), "-hostfile",
               self.args.hostfile, "--mca", "btl", "^openib",
               "--mca", "btl_tcp_if_include", "eth0"]
        for k, v in self.exports.items():
            cmd += ["-x", f"{k}={v}"]
        return cmd + [sys.executable, "-u"] + self.user_arguments


class MPICHRunner(MultiNodeRunner):
    name = "mpich"

    def backend_exists(self):
        return shutil.which("mpirun") is not None

    def get_cmd(self, environment=None):
        cmd = ["mpirun", "-n", str(self.nprocs),
               "-ppn", str(max(len(g) for g in self.world_info.values())),
               "-hostfile", self.args.hostfile]
        for k, v in self.exports.items():
            cmd += ["-genv", k, v]
        return cmd + [sys.executable, "-u"] + self.user_arguments


class IMPIRunner(MultiNodeRunner):
    name = "impi"

    def backend_exists(self):
        return shutil.which("mpirun") is not None

    def get_cmd(self, environment=None):
        ppn = max(len(g) for g in self.world_info.values())
        cmd = ["mpirun", "-ppn", str(ppn), "-n", str(self.nprocs),
               "-hostfile", self.args.hostfile]
        for k, v in self.exports.items():
            cmd += ["-genv", k, v]
        return cmd + [sys.executable, "-u"] + self.user_arguments


class SlurmRunner(MultiNodeRunner):
    name = "slurm"

    def backend_exists(self):
        return shutil.which("srun") is not None

    def get_cmd(self, environment=None):
        cmd = ["srun", "--nodes", str(self.nnodes),
               "--ntasks", str(self.nprocs),
               "--ntasks-per-node",
               str(max(len(g) for g in self.world_info.values()))]
        if self.exports:
            cmd += ["--export",
                    "ALL," + ",".join(f"{k}={v}"
                                      for k, v in self.exports.items())]
        return cmd + [sys.executable, "-u"] + self.user_arguments


class MVAPICHRunner(MultiNodeRunner):
    name = "mvapich"

    def backend_exists(self):
        return shutil.which("mpirun_rsh") is not None

    def get_cmd(self, environment=None):
        cmd = ["mpirun_rsh", "-np", str(self.nprocs),
               "-hostfile", self.args.hostfile]
        cmd += [f"{k}={v}" for k, v in self.exports.items()]
        return cmd + [sys.executable, "-u"] + self.user_arguments


RUNNERS = {r.name: r for r in (PDSHRunner, OpenMPIRunner, MPICHRunner,
                               IMPIRunner, SlurmRunner, MVAPICHRunner)}


def get_runner(name, args, world_info):
    try:
        return RUNNERS[name](args, world_info)
    except KeyError:
        raise ValueError(f"unknown launcher {name!r}; "
                         f"choices: {sorted(RUNNERS)} or ssh/local")
