#!/bin/bash
# DNS pre-resolution gate (same behavior as the reference's
# build/base/entrypoint.sh): before exec'ing the container command, wait
# until this pod's own FQDN and every host named in the hostfile resolve —
# launcher startup otherwise races the headless-Service DNS propagation and
# amdrun/mpirun fails its first ssh fan-out.
set -u

hostfile="${MPIAMD_HOSTFILE:-/etc/mpi/hostfile}"

resolve() {
    # getent covers /etc/hosts entries that nslookup misses
    getent hosts "$1" > /dev/null 2>&1 || nslookup "$1" > /dev/null 2>&1
}

wait_for() {
    local host="$1" tries=0
    until resolve "$host"; do
        tries=$((tries + 1))
        if [ "$tries" -gt 30 ]; then
            echo "entrypoint: $host still unresolved after $tries attempts" >&2
            return 1
        fi
        sleep "$(awk -v t="$tries" 'BEGIN { print (t < 10) ? t * 0.2 : 2 }')"
    done
}

me="$(hostname -f 2>/dev/null || hostname)"
wait_for "$me" || true

if [ -r "$hostfile" ]; then
    # hostfile lines: "<host> slots=N" (OpenMPI form) or "<host>:N"
    while read -r line; do
        host="${line%% *}"
        host="${host%%:*}"
        [ -n "$host" ] && wait_for "$host"
    done < "$hostfile"
fi

exec "$@"
