#!/bin/bash
# Generate the mutual-TLS CA + component keypairs (reference
# test/setup-ca.sh, openssl instead of certstrap) and print a
# Kubernetes secret manifest for them.
#
#   deploy/setup-ca.sh <output-dir> [name...]
#
# Default names cover one node with 8 GPU cards: component.registry,
# user.admin, and controller.<host>-gpu{0..7} / host.<host>-gpu{0..7}.
set -euo pipefail

DIR="${1:?usage: setup-ca.sh <output-dir> [name...]}"
shift || true
HOST="${OIM_HOST:-host-0}"
if [ $# -gt 0 ]; then
    NAMES=("$@")
else
    NAMES=(component.registry user.admin)
    for i in $(seq 0 7); do
        NAMES+=("controller.${HOST}-gpu${i}" "host.${HOST}-gpu${i}")
    done
fi

mkdir -p "$DIR"
if [ ! -f "$DIR/ca.key" ]; then
    openssl req -x509 -newkey rsa:2048 -nodes -keyout "$DIR/ca.key" \
        -out "$DIR/ca.crt" -days 365 -subj "/CN=OIM CA" 2>/dev/null
fi

for name in "${NAMES[@]}"; do
    [ -f "$DIR/$name.key" ] && continue
    openssl req -newkey rsa:2048 -nodes -keyout "$DIR/$name.key" \
        -out "$DIR/$name.csr" -subj "/CN=$name" 2>/dev/null
    printf 'subjectAltName=DNS:%s\n' "$name" > "$DIR/$name.ext"
    openssl x509 -req -in "$DIR/$name.csr" -CA "$DIR/ca.crt" \
        -CAkey "$DIR/ca.key" -CAcreateserial -out "$DIR/$name.crt" \
        -days 365 -extfile "$DIR/$name.ext" 2>/dev/null
    rm -f "$DIR/$name.csr" "$DIR/$name.ext"
done

# Emit a secret manifest (reference test/setup-ca.sh secret.yaml).
{
    echo "apiVersion: v1"
    echo "kind: Secret"
    echo "metadata:"
    echo "  name: oim-tls"
    echo "  namespace: oim"
    echo "data:"
    for f in "$DIR"/*.crt "$DIR"/*.key; do
        echo "  $(basename "$f"): $(base64 -w0 < "$f")"
    done
} > "$DIR/secret.yaml"
echo "wrote $DIR/secret.yaml (${#NAMES[@]} keypairs + CA)"
