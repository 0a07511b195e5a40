"""Drop-in compatibility: a state store populated BY THE REFERENCE (its
documented etcd value format, /root/reference/api/gpu-docker-api-en.md:365-515)
must be readable and operable by this daemon — info, history, patch, delete.
"""
import json

from gpu_docker_api_amd.models import GpuPatch, PatchRequest
from gpu_docker_api_amd.routers.app import Daemon
from gpu_docker_api_amd.state.keys import (
    CONTAINER_VERSION_MAP_KEY,
    Resource,
    resource_key,
)
from helpers import make_config

# A container spec exactly as the reference writes it (moby-typed JSON; shape
# from the documented example — Config/HostConfig/NetworkingConfig/Platform
# at the top level, camelCase wrapper keys, nvidia-era fields included).
REFERENCE_SPEC = {
    "version": 2,
    "createTime": "2024-01-22 07:36:04",
    "config": {
        "Hostname": "",
        "User": "",
        "AttachStdin": False,
        "ExposedPorts": {"22/tcp": {}},
        "Tty": True,
        "Env": ["USER=foo", "CONTAINER_VERSION=2"],
        "Cmd": None,
        "Image": "nvidia/cuda:10.0-base",
        "Volumes": None,
        "WorkingDir": "",
        "Entrypoint": None,
        "OnBuild": None,
        "Labels": None,
    },
    "hostConfig": {
        "Binds": ["veil-0:/root/veil-0"],
        "LogConfig": {"Type": "", "Config": None},
        "NetworkMode": "",
        "PortBindings": {"22/tcp": [{"HostIp": "", "HostPort": "40000"}]},
        "RestartPolicy": {"Name": "unless-stopped", "MaximumRetryCount": 0},
        "Runtime": "nvidia",
        "DeviceRequests": [
            {"Driver": "cdi", "DeviceIDs": ["nvidia.com/gpu=GPU-deadbeef"]}
        ],
        "Memory": 2147483648,
        "CpusetCpus": "0,1",
        "ShmSize": 274877906944,
        "StorageOpt": {"size": "30G"},
    },
    "networkingConfig": {"EndpointsConfig": {}},
    "platform": {"architecture": "", "os": ""},
    "containerName": "legacy-2",
}


def test_adopt_reference_written_state(tmp_path, run):
    async def main():
        d = Daemon(make_config(tmp_path))
        await d.start()
        # simulate the reference having written this deployment's state
        await d.store.put(
            Resource.CONTAINERS, "legacy", json.dumps(REFERENCE_SPEC, separators=(",", ":"))
        )
        await d.store.put(
            Resource.VERSIONS,
            CONTAINER_VERSION_MAP_KEY,
            json.dumps({"legacy": 2}),
        )
        await d.container_versions.load()

        # read path: info + history see the reference's spec untouched
        info = await d.replicaset.get_container_info("legacy")
        assert info["containerName"] == "legacy-2"
        assert info["config"]["Image"] == "nvidia/cuda:10.0-base"
        assert info["hostConfig"]["Binds"] == ["veil-0:/root/veil-0"]
        hist = await d.replicaset.get_container_history("legacy")
        assert hist[0]["status"]["version"] == 2

        # mutate path: patch performs a rolling replace on the adopted spec;
        # unknown moby fields round-trip; nvidia-era fields are inert here
        out = await d.replicaset.patch_container(
            "legacy", PatchRequest(gpu_patch=GpuPatch(gpu_count=1))
        )
        assert out["containerName"] == "legacy-3"
        new_info = await d.replicaset.get_container_info("legacy")
        assert new_info["version"] == 3
        # moby fields we never touch survive the round trip byte-for-byte
        assert new_info["config"]["Tty"] is True
        assert new_info["hostConfig"]["LogConfig"] == {"Type": "", "Config": None}
        # our GPU allocation recorded; the reference's stale CDI request kept
        # in the stored spec (drivers strip it at create time)
        st = await d.runtime.inspect("legacy-3")
        assert len(st.gpu_uuids) == 1

        await d.replicaset.delete_container("legacy")
        assert d.container_versions.get("legacy") is None
        await d.stop()

    run(main())
