"""Volume request DTOs (reference: internal/models/volume.go:26-39)."""
from __future__ import annotations

from pydantic import BaseModel


class VolumeCreate(BaseModel):
    name: str = ""
    size: str = ""


class VolumeSize(BaseModel):
    size: str = ""
