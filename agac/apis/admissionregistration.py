"""admissionregistration.k8s.io/v1 ValidatingWebhookConfiguration.

The reference never models this type in Go — it relies on a real
apiserver consuming ``config/webhook/manifests.yaml``.  Here the hermetic
apiserver consumes it too (``agac/kube/dynamicadmission.py``): applying
the manifest activates admission exactly like a cluster would, which is
what the reference's kind e2e exercises (``e2e/e2e_test.go:77-103``).
Field names follow the k8s API (camelCase on the wire via meta.to_dict).
"""

from __future__ import annotations

import typing
from dataclasses import dataclass, field

from .meta import ObjectMeta


@dataclass(slots=True)
class WebhookServiceReference:
    name: str = ""
    namespace: str = ""
    path: str = ""
    port: int = 443


@dataclass(slots=True)
class WebhookClientConfig:
    url: typing.Optional[str] = None
    service: typing.Optional[WebhookServiceReference] = None
    ca_bundle: typing.Optional[str] = None  # base64 PEM, like the real field


@dataclass(slots=True)
class RuleWithOperations:
    api_groups: typing.List[str] = field(default_factory=list)
    api_versions: typing.List[str] = field(default_factory=list)
    operations: typing.List[str] = field(default_factory=list)
    resources: typing.List[str] = field(default_factory=list)


@dataclass(slots=True)
class ValidatingWebhook:
    name: str = ""
    client_config: WebhookClientConfig = field(default_factory=WebhookClientConfig)
    rules: typing.List[RuleWithOperations] = field(default_factory=list)
    failure_policy: str = "Fail"
    side_effects: typing.Optional[str] = None
    admission_review_versions: typing.List[str] = field(default_factory=list)
    timeout_seconds: int = 10


@dataclass(slots=True)
class ValidatingWebhookConfiguration:
    kind: typing.ClassVar[str] = "ValidatingWebhookConfiguration"
    api_version: typing.ClassVar[str] = "admissionregistration.k8s.io/v1"

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    webhooks: typing.List[ValidatingWebhook] = field(default_factory=list)
