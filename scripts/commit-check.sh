#!/usr/bin/env bash
# Reject obviously malformed commit subjects on the latest commit
# (the analog of the reference's commit-check-latest.sh).
set -euo pipefail

subject=$(git log -1 --pretty=%s)

if [ -z "$subject" ]; then
    echo "error: empty commit subject" >&2
    exit 1
fi

if [ "${#subject}" -gt 100 ]; then
    echo "error: commit subject longer than 100 characters" >&2
    exit 1
fi

case "$subject" in
    fixup!*|squash!*|WIP*|wip*)
        echo "error: unfinished commit subject: $subject" >&2
        exit 1
        ;;
esac

echo "commit subject ok: $subject"
