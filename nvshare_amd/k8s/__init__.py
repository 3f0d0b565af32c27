"""Kubernetes integration: device plugin (kubelet v1beta1) + manifests."""
