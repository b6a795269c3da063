"""
Pydantic models of the k8s config fragments accepted in Machine
runtime sections (spec: gordo/workflow/config_elements/schemas.py).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from pydantic import BaseModel, ConfigDict


class EnvVar(BaseModel):
    name: str
    value: Optional[str] = None
    valueFrom: Optional[Dict[str, Any]] = None


class CSIVolumeSource(BaseModel):
    driver: str
    readOnly: Optional[bool] = None
    volumeAttributes: Optional[Dict[str, str]] = None


class Volume(BaseModel):
    name: str
    csi: Optional[CSIVolumeSource] = None
    persistentVolumeClaim: Optional[Dict[str, Any]] = None
    emptyDir: Optional[Dict[str, Any]] = None
    configMap: Optional[Dict[str, Any]] = None
    secret: Optional[Dict[str, Any]] = None


class VolumeMount(BaseModel):
    name: str
    mountPath: str
    readOnly: Optional[bool] = None
    subPath: Optional[str] = None


class ResourceRequirements(BaseModel):
    requests: Optional[Dict[str, Any]] = None
    limits: Optional[Dict[str, Any]] = None


class PodRuntime(BaseModel):
    image: Optional[str] = None
    resources: Optional[ResourceRequirements] = None

    model_config = ConfigDict(extra="allow")


class RemoteLogging(BaseModel):
    enable: bool = False


class BuilderPodRuntime(PodRuntime):
    remote_logging: Optional[RemoteLogging] = None
    env: Optional[List[EnvVar]] = None


class SecurityContext(BaseModel):
    runAsUser: Optional[int] = None
    runAsGroup: Optional[int] = None
    runAsNonRoot: Optional[bool] = None
    allowPrivilegeEscalation: Optional[bool] = None

    model_config = ConfigDict(extra="allow")


class PodSecurityContext(SecurityContext):
    fsGroup: Optional[int] = None
    supplementalGroups: Optional[List[int]] = None
