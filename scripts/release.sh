#!/usr/bin/env bash
# Release packaging — the analog of the reference's .goreleaser.yml
# (multi-OS binaries + completions packaging + version stamping,
# reference .goreleaser.yml:6-60).  Python wheels are platform-neutral,
# so one wheel + sdist replaces the per-OS binary matrix; completions
# are generated from the CLI itself and archived alongside.
set -euo pipefail

cd "$(dirname "$0")/.."
VERSION=$(python -c "import operator_builder_amd; print(operator_builder_amd.__version__)")
OUT=dist/release-$VERSION
rm -rf "$OUT" && mkdir -p "$OUT/completions"

# distributables (sdist + wheel when the build backend is available)
if python -c "import build" 2>/dev/null; then
    python -m build --outdir "$OUT"
else
    python setup.py -q sdist --dist-dir "$OUT"
fi

# shell completions (reference goreleaser packages bash/zsh/fish)
for shell in bash zsh fish; do
    python -m operator_builder_amd.cli.main completion "$shell" \
        > "$OUT/completions/operator-builder.$shell"
done

tar -czf "$OUT/completions.tar.gz" -C "$OUT" completions
echo "release artifacts in $OUT:"
ls -l "$OUT"
