"""agac — a clean-room Kubernetes controller framework with the capabilities of
h3poteto/aws-global-accelerator-controller.

The reference is a pure-Go, network-I/O-bound Kubernetes controller (see
/root/repo/SURVEY.md); per BASELINE.json this build is tiered as a k8s
controller (no GPU surface).  agac re-implements the whole stack in Python:

- ``agac.kube``        — API machinery: an in-memory API server with watch
                         semantics, typed clients, shared informers/listers,
                         rate-limited workqueues, event recording and
                         lease-based leader election (replaces client-go and
                         the generated clientset, reference ``pkg/client/``).
- ``agac.reconcile``   — the generic reconcile engine
                         (reference ``pkg/reconcile/reconcile.go``).
- ``agac.cloudprovider`` — the cloud-provider seam: AWS resource managers for
                         Global Accelerator / ELBv2 / Route53 plus a stateful
                         in-memory AWS fake (reference ``pkg/cloudprovider``).
- ``agac.controller``  — the three controllers: GlobalAccelerator, Route53 and
                         EndpointGroupBinding (reference ``pkg/controller``).
- ``agac.webhook``     — validating admission webhook for the
                         EndpointGroupBinding CRD (reference ``pkg/webhoook``).
- ``agac.manager``     — controller manager wiring
                         (reference ``pkg/manager/manager.go``).
"""

__version__ = "0.1.0"
