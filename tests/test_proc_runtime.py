"""ProcRuntime: real supervised processes, no GPU required (GPU visibility
injection is covered by test_gpu.py on the MI355X box)."""
import os
import signal
import time

from gpu_docker_api_amd.models.etcd import ContainerSpec
from gpu_docker_api_amd.runtime.proc import ProcRuntime


def _spec(name, cmd=None, env=None, image=""):
    s = ContainerSpec()
    s.config = {"Image": image, "Env": list(env or []), "Cmd": list(cmd or [])}
    s.host_config = {}
    s.container_name = name
    return s


def test_lifecycle_real_process(tmp_path, run):
    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        await rt.create(_spec("c-1", cmd=["sleep", "30"]))
        st = await rt.inspect("c-1")
        assert st is not None and not st.running
        await rt.start("c-1")
        st = await rt.inspect("c-1")
        assert st.running and st.pid > 0
        os.kill(st.pid, 0)  # process really exists
        await rt.stop("c-1")
        st = await rt.inspect("c-1")
        assert not st.running and st.pid == 0
        await rt.start("c-1")
        assert (await rt.inspect("c-1")).running
        await rt.remove("c-1", force=True)
        assert await rt.inspect("c-1") is None
        await rt.close()

    run(main())


def test_exited_process_detected(tmp_path, run):
    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        await rt.create(_spec("c-1", cmd=["sh", "-c", "exit 0"]))
        await rt.start("c-1")
        for _ in range(100):
            st = await rt.inspect("c-1")
            if not st.running:
                break
            time.sleep(0.05)
        assert not st.running and st.status == "exited"
        await rt.close()

    run(main())


def test_restart_policy_supervisor(tmp_path, run):
    import asyncio

    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        spec = _spec("c-1", cmd=["sh", "-c", "sleep 0.2"])
        spec.host_config["RestartPolicy"] = {"Name": "unless-stopped"}
        await rt.create(spec)
        await rt.start("c-1")
        first_pid = (await rt.inspect("c-1")).pid
        # process exits after 0.2 s; supervisor must bring it back
        restarted = False
        for _ in range(60):
            await asyncio.sleep(0.1)
            st = await rt.inspect("c-1")
            if st.running and st.pid != first_pid:
                restarted = True
                break
        assert restarted, "supervisor did not restart the exited container"
        # manual stop must NOT be resurrected
        await rt.stop("c-1")
        await asyncio.sleep(0.8)
        assert not (await rt.inspect("c-1")).running
        await rt.close()

    run(main())


def test_exec_runs_in_rootfs_with_env(tmp_path, run):
    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        await rt.create(_spec("c-1", cmd=["sleep", "30"], env=["FOO=bar42"]))
        await rt.start("c-1")
        out = await rt.execute("c-1", ["sh", "-c", "echo $FOO; pwd"])
        assert "bar42" in out
        st = await rt.inspect("c-1")
        assert st.upper_dir in out
        # workdir
        out = await rt.execute("c-1", ["pwd"], workdir="/sub/dir")
        assert out.strip().endswith("sub/dir")
        await rt.close()

    run(main())


def test_pause_unpause_signals(tmp_path, run):
    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        await rt.create(_spec("c-1", cmd=["sleep", "30"]))
        await rt.start("c-1")
        import asyncio

        st = await rt.inspect("c-1")

        async def wait_state(pid, want, timeout=3.0):
            # signal delivery is asynchronous: poll /proc until it lands
            deadline = asyncio.get_event_loop().time() + timeout
            state = "?"
            while asyncio.get_event_loop().time() < deadline:
                with open(f"/proc/{pid}/stat") as f:
                    state = f.read().split(") ")[1].split()[0]
                if state in want:
                    return state
                await asyncio.sleep(0.02)
            return state

        await rt.pause("c-1")
        state = await wait_state(st.pid, ("T", "t"))
        assert state in ("T", "t"), f"expected stopped, got {state}"
        assert (await rt.inspect("c-1")).paused
        await rt.unpause("c-1")
        state = await wait_state(st.pid, ("S", "R"))
        assert state in ("S", "R"), f"expected running, got {state}"
        await rt.close()

    run(main())


def test_commit_and_image_seed(tmp_path, run):
    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        await rt.create(_spec("c-1", cmd=["sleep", "30"]))
        await rt.start("c-1")
        st = await rt.inspect("c-1")
        with open(os.path.join(st.upper_dir, "state.txt"), "w") as f:
            f.write("snapshot-me")
        ref = await rt.commit("c-1", "myimg", "v1")
        assert ref == "myimg:v1"
        # new container from the committed image inherits the rootfs
        await rt.create(_spec("c2-1", cmd=["sleep", "30"], image="myimg:v1"))
        st2 = await rt.inspect("c2-1")
        assert open(os.path.join(st2.upper_dir, "state.txt")).read() == "snapshot-me"
        await rt.close()

    run(main())


def test_volume_bind_symlink(tmp_path, run):
    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        vol = await rt.volume_create("v-1", {"size": "1GB"})
        with open(os.path.join(vol.mountpoint, "data.txt"), "w") as f:
            f.write("vol-data")
        spec = _spec("c-1", cmd=["sleep", "30"])
        spec.host_config["Binds"] = ["v-1:/data"]
        await rt.create(spec)
        await rt.start("c-1")
        out = await rt.execute("c-1", ["cat", "data.txt"], workdir="/data")
        assert "vol-data" in out
        # volumes survive runtime restart (opts.json reload)
        rt2 = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        v = await rt2.volume_inspect("v-1")
        assert v is not None and v.options.get("size") == "1GB"
        await rt.close()

    run(main())


def test_stop_kills_whole_process_group(tmp_path, run):
    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        # parent spawns a child; stop must kill both (exact pgid, no patterns)
        await rt.create(_spec("c-1", cmd=["sh", "-c", "sleep 60 & wait"]))
        await rt.start("c-1")
        st = await rt.inspect("c-1")
        pgid = os.getpgid(st.pid)
        await rt.stop("c-1", timeout=3)

        def live_members():
            # count non-zombie members of the group (orphaned zombies keep
            # the pgid registered until init reaps them — not "alive")
            n = 0
            for pid in os.listdir("/proc"):
                if not pid.isdigit():
                    continue
                try:
                    if os.getpgid(int(pid)) != pgid:
                        continue
                    with open(f"/proc/{pid}/stat") as f:
                        state = f.read().split(") ")[1].split()[0]
                    if state not in ("Z", "X"):
                        n += 1
                except (OSError, IndexError):
                    continue
            return n

        for _ in range(100):
            if live_members() == 0:
                break
            time.sleep(0.02)
        assert live_members() == 0, "process group survived stop"
        await rt.close()

    run(main())


def test_stop_kills_paused_container(tmp_path, run):
    """SIGTERM stays pending on a SIGSTOPped group — stop must SIGCONT it
    (found as hours-old frozen sleeps leaked by earlier test runs)."""
    import asyncio

    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        await rt.create(_spec("c-1", cmd=["sleep", "30"]))
        await rt.start("c-1")
        await rt.pause("c-1")
        st = await rt.inspect("c-1")
        pid = st.pid
        await rt.stop("c-1", timeout=3)
        for _ in range(100):
            try:
                os.kill(pid, 0)
            except ProcessLookupError:
                break
            await asyncio.sleep(0.02)
        else:
            # zombie state also counts as dead for this purpose
            with open(f"/proc/{pid}/stat") as f:
                assert f.read().split(") ")[1].split()[0] == "Z"
        await rt.close()

    run(main())


def test_daemon_restart_adopts_running_container(tmp_path, run):
    """dockerd containers survive the daemon; the proc runtime must too:
    a new runtime instance on the same data dir adopts the still-running
    process (pid + /proc starttime identity) and can stop it."""
    import os

    from gpu_docker_api_amd.models.etcd import ContainerSpec
    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    async def main():
        rt1 = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        spec = ContainerSpec()
        spec.config = {"Cmd": ["sleep", "60"]}
        spec.container_name = "adopt-1"
        await rt1.create(spec)
        await rt1.start("adopt-1")
        pid = (await rt1.inspect("adopt-1")).pid
        assert pid > 0
        # simulate daemon death: drop the runtime WITHOUT stopping anything
        if rt1._supervisor is not None:
            rt1._supervisor.cancel()

        rt2 = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        st = await rt2.inspect("adopt-1")
        assert st is not None and st.running and st.pid == pid
        assert os.path.exists(f"/proc/{pid}")
        stats = await rt2.stats("adopt-1")
        assert stats["running"] and stats["memoryBytes"] > 0
        # the adopted process is stoppable through the new runtime
        await rt2.stop("adopt-1", timeout=5)
        assert not os.path.exists(f"/proc/{pid}") or open(f"/proc/{pid}/stat").read().split()[2] == "Z"
        st = await rt2.inspect("adopt-1")
        assert not st.running
        await rt2.remove("adopt-1", force=True)
        await rt2.close()

    run(main())


def test_daemon_restart_resurrects_dead_unless_stopped(tmp_path, run):
    """A container that died while the daemon was down is restarted by the
    new daemon's supervisor when its policy says unless-stopped (docker's
    restart-on-daemon-start semantics)."""
    import asyncio
    import os
    import signal as _signal

    from gpu_docker_api_amd.models.etcd import ContainerSpec
    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    async def main():
        rt1 = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        spec = ContainerSpec()
        spec.config = {"Cmd": ["sleep", "60"]}
        spec.host_config = {"RestartPolicy": {"Name": "unless-stopped"}}
        spec.container_name = "res-1"
        await rt1.create(spec)
        await rt1.start("res-1")
        pid = (await rt1.inspect("res-1")).pid
        if rt1._supervisor is not None:
            rt1._supervisor.cancel()
        # kill the workload while "no daemon" is watching (it stays a
        # zombie of THIS process until reaped — the runtime must treat
        # state Z as dead)
        os.killpg(pid, _signal.SIGKILL)
        await asyncio.sleep(0.1)

        rt2 = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        # the new supervisor must resurrect it
        deadline = asyncio.get_event_loop().time() + 5
        st = await rt2.inspect("res-1")
        while asyncio.get_event_loop().time() < deadline and not st.running:
            await asyncio.sleep(0.1)
            st = await rt2.inspect("res-1")
        assert st.running and st.pid != pid
        await rt2.remove("res-1", force=True)
        await rt2.close()

    run(main())


def test_full_daemon_restart_with_live_workload(tmp_path, run):
    """Daemon-level: a proc-runtime replicaSet keeps RUNNING through a
    daemon crash; the next daemon adopts it (runtime metadata) AND still
    knows its versions/allocations (persisted store) — then can exec into
    it, patch it, and delete it."""
    import os

    from gpu_docker_api_amd.models import ContainerExecute, ContainerRun, GpuPatch, PatchRequest
    from gpu_docker_api_amd.routers.app import Daemon
    from helpers import make_config

    async def main():
        cfg = make_config(tmp_path, runtime="proc")
        d1 = Daemon(cfg)
        await d1.start()
        await d1.replicaset.run_gpu_container(
            ContainerRun(image_name="img", replica_set_name="live", gpu_count=1,
                         cmd=["sleep", "120"])
        )
        pid = (await d1.runtime.inspect("live-1")).pid
        await d1.queue.close()  # flush write-behind state; no graceful stop

        d2 = Daemon(make_config(tmp_path, runtime="proc"), store=d1.store)
        await d2.start()
        st = await d2.runtime.inspect("live-1")
        assert st is not None and st.running and st.pid == pid  # adopted
        assert sum(d2.gpu.get_gpu_status().values()) == 1       # still allocated
        out, rc = await d2.replicaset.execute_container(
            "live", ContainerExecute(cmd=["sh", "-c", "echo adopted-$PPID"])
        )
        assert rc == 0 and "adopted-" in out
        res = await d2.replicaset.patch_container(
            "live", PatchRequest(gpu_patch=GpuPatch(gpu_count=0))
        )
        assert res["containerName"] == "live-2"
        assert not os.path.exists(f"/proc/{pid}") or \
            open(f"/proc/{pid}/stat").read().split()[2] == "Z"  # old one stopped
        await d2.replicaset.delete_container("live")
        await d2.stop()

    run(main())


def test_crash_leftover_deleting_dirs_swept(tmp_path, run):
    """A crash between the delete-rename and the batched rm leaves a
    '<name>.deleting-*' dir; the next runtime sweeps it at load."""
    import os

    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    leftover = tmp_path / "trash" / "old-1-deadbeef"
    leftover.mkdir(parents=True)
    (leftover / "junk.bin").write_bytes(b"x" * 128)
    legacy = tmp_path / "containers" / "old-2.deleting-cafe"
    legacy.mkdir(parents=True)

    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        assert not os.path.exists(leftover)
        assert not os.path.exists(legacy)
        await rt.close()

    run(main())


def test_crash_loop_backoff(tmp_path, run):
    """A workload that dies instantly must not be restarted at full
    supervisor tick rate: the restart counter must grow slower than the
    tick rate (docker's exponential crash-loop backoff)."""
    import asyncio

    from gpu_docker_api_amd.models.etcd import ContainerSpec
    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    async def main():
        rt = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        spec = ContainerSpec()
        spec.config = {"Cmd": ["false"]}  # exits immediately, rc 1
        spec.host_config = {"RestartPolicy": {"Name": "always"}}
        spec.container_name = "loop-1"
        await rt.create(spec)
        await rt.start("loop-1")
        await asyncio.sleep(2.0)
        p = rt._procs["loop-1"]
        # ~10 ticks/2s without backoff would mean ~10 restarts; with
        # 0.1*2^n backoff the schedule is 0.2+0.4+0.8+... -> at most 4-5
        assert 1 <= p.restarts <= 6, p.restarts
        first = p.restarts
        await asyncio.sleep(1.5)
        assert p.restarts - first <= 2  # slowing down, not tick-rate
        await rt.remove("loop-1", force=True)
        await rt.close()

    run(main())


def test_adopted_paused_container_shows_paused(tmp_path, run):
    """A container paused (SIGSTOP) before the daemon died must adopt as
    paused, and unpause must resume it."""
    from gpu_docker_api_amd.models.etcd import ContainerSpec
    from gpu_docker_api_amd.runtime.proc import ProcRuntime

    async def main():
        rt1 = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        spec = ContainerSpec()
        spec.config = {"Cmd": ["sleep", "60"]}
        spec.container_name = "pz-1"
        await rt1.create(spec)
        await rt1.start("pz-1")
        await rt1.pause("pz-1")
        if rt1._supervisor is not None:
            rt1._supervisor.cancel()

        rt2 = ProcRuntime(base_dir=str(tmp_path), use_cgroups=False)
        st = await rt2.inspect("pz-1")
        assert st is not None and st.paused and st.status == "paused"
        await rt2.unpause("pz-1")
        st = await rt2.inspect("pz-1")
        assert not st.paused and st.running
        await rt2.remove("pz-1", force=True)
        await rt2.close()

    run(main())
