#!/usr/bin/env bash
# Install `kubectl notebook` / `kubectl applybuild` as thin wrappers over
# the `sub` CLI (parity: reference install/kubectl-plugins.sh).
set -euo pipefail
BIN=${BIN:-/usr/local/bin}
cat > "$BIN/kubectl-notebook" <<'SH'
#!/usr/bin/env bash
exec sub notebook "$@"
SH
cat > "$BIN/kubectl-applybuild" <<'SH'
#!/usr/bin/env bash
exec sub apply "$@"
SH
chmod +x "$BIN/kubectl-notebook" "$BIN/kubectl-applybuild"
echo "installed kubectl-notebook, kubectl-applybuild to $BIN"
