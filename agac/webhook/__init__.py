"""Validating admission webhook for the EndpointGroupBinding CRD
(reference ``pkg/webhoook/`` — sic, the reference directory name has a
typo; this package does not reproduce it)."""
