# Pinned versions for image builds and CI (single source of truth,
# same role as the reference's versions.mk).

VERSION ?= v0.1.0
ROCM_VERSION ?= 7.2
ROCM_IMAGE ?= rocm/dev-ubuntu-22.04:$(ROCM_VERSION)
DISTROLESS_IMAGE ?= gcr.io/distroless/python3-debian12:nonroot
GPU_ARCH ?= gfx950

REGISTRY ?= ghcr.io/example
IMAGE_NAME ?= k8s-cc-manager-amd
IMAGE ?= $(REGISTRY)/$(IMAGE_NAME)
