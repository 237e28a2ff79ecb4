#!/bin/bash
# Build + tag the deployment images (reference hack/update-images.sh).
set -euo pipefail
cd "$(dirname "$0")/.."
TAG="${1:-latest}"
docker build -f build/docker/Dockerfile.hipstored -t "oim-amd/hipstored:${TAG}" .
docker build -f build/docker/Dockerfile.control-plane -t "oim-amd/oim-csi-driver:${TAG}" .
docker tag "oim-amd/oim-csi-driver:${TAG}" "oim-amd/oim-registry:${TAG}"
echo "built oim-amd/hipstored:${TAG} oim-amd/oim-csi-driver:${TAG}"
