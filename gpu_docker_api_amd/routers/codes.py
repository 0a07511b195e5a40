"""Numeric business codes — the wire contract clients depend on.

Values and messages reproduce the reference's table exactly
(/root/reference/internal/routers/code.go:5-97); every response is HTTP 200
with ``{code, msg, data}`` (response.go:9-29).
"""
from __future__ import annotations

import enum


class Code(int, enum.Enum):
    SUCCESS = 200
    SERVER_BUSY = 500
    FORBIDDEN = 403

    INVALID_PARAMS = 1000
    IMAGE_NAME_EMPTY = 1001
    CONTAINER_NAME_EMPTY = 1002
    CONTAINER_NAME_DASH = 1003
    CONTAINER_RUN_FAILED = 1004
    CONTAINER_DELETE_FAILED = 1005
    CONTAINER_EXECUTE_FAILED = 1006
    CONTAINER_PATCH_FAILED = 1007
    CONTAINER_ALREADY_EXIST = 1008
    CONTAINER_NO_NEED_PATCH = 1009
    CONTAINER_STOP_FAILED = 1010
    CONTAINER_RESTART_FAILED = 1011
    GPU_COUNT_GE_ZERO = 1012
    CONTAINER_GPU_NOT_ENOUGH = 1013
    CONTAINER_PORT_NOT_ENOUGH = 1014
    CONTAINER_COMMIT_FAILED = 1015
    CONTAINER_GET_INFO_FAILED = 1016
    CONTAINER_GET_HISTORY_FAILED = 1017
    CONTAINER_SHUTDOWN_FAILED = 1018
    CONTAINER_STARTUP_FAILED = 1019
    CONTAINER_VERSION_GE_ZERO = 1020
    CONTAINER_ROLLBACK_FAILED = 1021
    CONTAINER_NO_NEED_ROLLBACK = 1022
    CONTAINER_CPU_NOT_ENOUGH = 1023
    CPU_COUNT_GE_ZERO = 1024
    CONTAINER_MEMORY_UNIT = 1025

    VOLUME_CREATE_FAILED = 1100
    VOLUME_NAME_EMPTY = 1101
    VOLUME_DELETE_FAILED = 1102
    VOLUME_EXISTED = 1103
    VOLUME_NAME_MUST_CONTAIN_VERSION = 1104
    VOLUME_SIZE_NO_NEED_PATCH = 1105
    VOLUME_SIZE_UNIT = 1106
    VOLUME_SIZE_USED_GT_REDUCE = 1107
    VOLUME_NAME_DASH = 1108
    VOLUME_NAME_SLASH = 1109
    VOLUME_GET_INFO_FAILED = 1110
    VOLUME_GET_HISTORY_FAILED = 1111
    VOLUME_PATCH_FAILED = 1112


_MESSAGES = {
    Code.SUCCESS: "Success",
    Code.SERVER_BUSY: "Server busy",
    Code.FORBIDDEN: "Forbidden",
    Code.INVALID_PARAMS: "Failed to parse body",
    Code.IMAGE_NAME_EMPTY: "Image name cannot be empty",
    Code.CONTAINER_NAME_EMPTY: "Container name cannot be empty",
    Code.CONTAINER_NAME_DASH: "Container name cannot contain dash",
    Code.CONTAINER_RUN_FAILED: "Failed to start container",
    Code.CONTAINER_DELETE_FAILED: "Failed to delete container",
    Code.CONTAINER_EXECUTE_FAILED: "Failed to execute a command",
    Code.CONTAINER_PATCH_FAILED: "Failed to patch container",
    Code.CONTAINER_ALREADY_EXIST: "Container already exists",
    Code.CONTAINER_NO_NEED_PATCH: "Container doesn't need patch",
    Code.CONTAINER_STOP_FAILED: "Failed to stop container",
    Code.CONTAINER_RESTART_FAILED: "Failed to restart container",
    Code.GPU_COUNT_GE_ZERO: "GPU count must be greater than or equal to 0",
    Code.CONTAINER_GPU_NOT_ENOUGH: "Not enough GPU resources",
    Code.CONTAINER_PORT_NOT_ENOUGH: "Not enough port resources",
    Code.CONTAINER_COMMIT_FAILED: "Failed to commit image",
    Code.CONTAINER_GET_INFO_FAILED: "Failed to get container info, container not found",
    Code.CONTAINER_GET_HISTORY_FAILED: "Failed to get container history, container not found",
    Code.CONTAINER_SHUTDOWN_FAILED: "Failed to shut down container",
    Code.CONTAINER_STARTUP_FAILED: "Failed to start up container",
    Code.CONTAINER_VERSION_GE_ZERO: "Container version must be greater than or equal to 0",
    Code.CONTAINER_ROLLBACK_FAILED: "Failed to rollback container",
    Code.CONTAINER_NO_NEED_ROLLBACK: "Container doesn't need rollback, the current version is the same as the requested version",
    Code.CONTAINER_CPU_NOT_ENOUGH: "Not enough CPU resources",
    Code.CPU_COUNT_GE_ZERO: "CPU count must be greater than or equal to 0",
    Code.CONTAINER_MEMORY_UNIT: "Memory size units are not supported, supported units: KB, MB, GB, TB",
    Code.VOLUME_CREATE_FAILED: "Failed to create volume",
    Code.VOLUME_NAME_EMPTY: "Volume name cannot be empty",
    Code.VOLUME_DELETE_FAILED: "Failed to delete volume",
    Code.VOLUME_EXISTED: "Volume already exists",
    Code.VOLUME_NAME_MUST_CONTAIN_VERSION: "Volume name must contain the version number",
    Code.VOLUME_SIZE_NO_NEED_PATCH: "Volume doesn't need patch, as it is the same size before and after the update",
    Code.VOLUME_SIZE_UNIT: "Volume size units are not supported, supported units: KB, MB, GB, TB",
    Code.VOLUME_SIZE_USED_GT_REDUCE: "Failed to patch volume size, the patch size is smaller than the used size",
    Code.VOLUME_NAME_DASH: "Volume name cannot contain dash",
    Code.VOLUME_NAME_SLASH: "Volume name must not begin with /",
    Code.VOLUME_GET_INFO_FAILED: "Failed to get volume info",
    Code.VOLUME_GET_HISTORY_FAILED: "Failed to get volume history",
    Code.VOLUME_PATCH_FAILED: "Failed to patch volume",
}


def msg(code: Code) -> str:
    return _MESSAGES.get(code, _MESSAGES[Code.SERVER_BUSY])
