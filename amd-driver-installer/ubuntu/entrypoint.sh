#!/bin/bash
# amdgpu driver + ROCm userspace installer for Ubuntu nodes.
#
# Role parity: /root/reference/nvidia-driver-installer/ubuntu/entrypoint.sh
# (cache check keyed on kernel+driver version :33-51, kernel headers :70-74,
# installer run :122-135, verify :149-156, host ld.so.conf :158-163).
# MI355X redesign: no proprietary .run installer and no overlayfs trick —
# amdgpu-dkms builds against the node kernel from the ROCm apt repo, and the
# ROCm userspace the device plugin mounts into pods (hip runtime, rocm-smi,
# libamd_smi, librccl, rocBLAS/hipBLASLt/MIOpen) is copied into
# ROOT_MOUNT_DIR/home/kubernetes/bin/amd.
set -o errexit
set -o pipefail
set -u

set -x

ROCM_VERSION="${ROCM_VERSION:-7.2}"
AMDGPU_DRIVER_VERSION="${AMDGPU_DRIVER_VERSION:-30.20}"
# COS DaemonSets use the single knob INSTALL_PRELOADED_ONLY=true: the KMD
# is baked into the node image and COS has no apt — alias it onto the two
# fine-grained flags (the image's /opt/rocm is the userspace source).
if [[ "${INSTALL_PRELOADED_ONLY:-false}" == "true" ]]; then
  SKIP_KMD_BUILD=1
  SKIP_PACKAGE_INSTALL=1
fi
# install prefix of the ROCm userspace to stage from (versioned prefixes
# like /opt/rocm-7.2.0 exist on some images)
ROCM_DIR="${ROCM_DIR:-/opt/rocm}"
ROOT_MOUNT_DIR="${ROOT_MOUNT_DIR:-/root}"
INSTALL_DIR_HOST="${INSTALL_DIR_HOST:-/home/kubernetes/bin/amd}"
INSTALL_DIR="${ROOT_MOUNT_DIR}${INSTALL_DIR_HOST}"
CACHE_FILE="${INSTALL_DIR}/.cache"
KERNEL_VERSION="$(uname -r)"

check_cached_version() {
  # Parity: cache keyed on (kernel, driver version), entrypoint.sh:33-51.
  [[ -f "${CACHE_FILE}" ]] || return 1
  grep -q "^CACHE_KERNEL_VERSION=${KERNEL_VERSION}$" "${CACHE_FILE}" || return 1
  grep -q "^CACHE_AMDGPU_VERSION=${AMDGPU_DRIVER_VERSION}$" "${CACHE_FILE}" || return 1
  grep -q "^CACHE_ROCM_VERSION=${ROCM_VERSION}$" "${CACHE_FILE}" || return 1
  echo "amdgpu ${AMDGPU_DRIVER_VERSION} + ROCm ${ROCM_VERSION} already installed for ${KERNEL_VERSION}"
  return 0
}

update_cached_version() {
  cat >"${CACHE_FILE}" <<EOF
CACHE_KERNEL_VERSION=${KERNEL_VERSION}
CACHE_AMDGPU_VERSION=${AMDGPU_DRIVER_VERSION}
CACHE_ROCM_VERSION=${ROCM_VERSION}
EOF
}

install_kernel_headers() {
  # Parity: entrypoint.sh:70-74.
  apt-get update
  apt-get install -y "linux-headers-${KERNEL_VERSION}" || \
    apt-get install -y "linux-headers-generic"
}

setup_repos() {
  local ub_codename
  ub_codename="$(. /etc/os-release && echo "${VERSION_CODENAME}")"
  mkdir -p /etc/apt/keyrings
  curl -fsSL https://repo.radeon.com/rocm/rocm.gpg.key | \
    gpg --dearmor -o /etc/apt/keyrings/rocm.gpg
  echo "deb [arch=amd64 signed-by=/etc/apt/keyrings/rocm.gpg] \
https://repo.radeon.com/amdgpu/${AMDGPU_DRIVER_VERSION}/ubuntu ${ub_codename} main" \
    > /etc/apt/sources.list.d/amdgpu.list
  echo "deb [arch=amd64 signed-by=/etc/apt/keyrings/rocm.gpg] \
https://repo.radeon.com/rocm/apt/${ROCM_VERSION} ${ub_codename} main" \
    > /etc/apt/sources.list.d/rocm.list
  apt-get update
}

build_and_load_kmd() {
  # dkms builds the amdgpu kernel module against the node kernel.
  DEBIAN_FRONTEND=noninteractive apt-get install -y "amdgpu-dkms"
  dkms autoinstall -k "${KERNEL_VERSION}"
  modprobe amdgpu
}

install_rocm_userspace() {
  if [[ "${SKIP_PACKAGE_INSTALL:-0}" != "1" ]]; then
    DEBIAN_FRONTEND=noninteractive apt-get install -y \
      rocm-core rocminfo rocm-smi-lib amd-smi-lib \
      hip-runtime-amd rocblas hipblaslt miopen-hip rccl
  fi
  # SKIP_PACKAGE_INSTALL=1: the installer image already ships the ROCm
  # userspace at /opt/rocm (preloaded installer images, air-gapped
  # clusters); stage it directly without touching apt.
  mkdir -p "${INSTALL_DIR}/lib64" "${INSTALL_DIR}/bin" "${INSTALL_DIR}/.info"
  # The subset GPU pods need, mounted read-only by the device plugin
  # (manager.mounts(): /home/kubernetes/bin/amd -> /usr/local/amd).
  cp -a "${ROCM_DIR}"/lib/*.so* "${INSTALL_DIR}/lib64/" 2>/dev/null || true
  # ROCm binaries carry RUNPATH $ORIGIN/../lib — a lib->lib64 symlink
  # makes the staged tree self-contained for them
  ln -sfn lib64 "${INSTALL_DIR}/lib"
  for tool in rocminfo rocm-smi amd-smi hipconfig; do
    [[ -x "${ROCM_DIR}/bin/${tool}" ]] && cp -a "${ROCM_DIR}/bin/${tool}" "${INSTALL_DIR}/bin/"
  done
  # amd-smi is a symlink to ../libexec/amdsmi_cli/amdsmi_cli.py (a Python
  # CLI): stage libexec too or the staged bin/amd-smi dangles
  if [[ -d "${ROCM_DIR}/libexec/amdsmi_cli" ]]; then
    mkdir -p "${INSTALL_DIR}/libexec"
    cp -a "${ROCM_DIR}/libexec/amdsmi_cli" "${INSTALL_DIR}/libexec/"
  fi
  # ... and the amdsmi python package the CLI imports
  if [[ -d "${ROCM_DIR}/share/amd_smi/amdsmi" ]]; then
    mkdir -p "${INSTALL_DIR}/share/amd_smi"
    cp -a "${ROCM_DIR}/share/amd_smi/amdsmi" "${INSTALL_DIR}/share/amd_smi/"
  fi
  cp -a "${ROCM_DIR}/.info/version" "${INSTALL_DIR}/.info/version" 2>/dev/null || true
}

verify_installation() {
  # Parity: nvidia-smi verify + device-node check, entrypoint.sh:149-156.
  [[ -e "${ROOT_MOUNT_DIR}/dev/kfd" ]] || { echo "/dev/kfd missing"; return 1; }
  ls "${ROOT_MOUNT_DIR}"/dev/dri/renderD* >/dev/null
  # plain grep (not -q): -q exits at first match and SIGPIPEs rocminfo,
  # which pipefail turns into rc 141
  "${INSTALL_DIR}/bin/rocminfo" | grep "gfx950" > /dev/null
  "${INSTALL_DIR}/bin/amd-smi" list
}

update_host_ld_cache() {
  # Parity: entrypoint.sh:158-163.
  echo "${INSTALL_DIR_HOST}/lib64" > "${ROOT_MOUNT_DIR}/etc/ld.so.conf.d/amd.conf"
  ldconfig -r "${ROOT_MOUNT_DIR}"
}

main() {
  if check_cached_version; then
    verify_installation
    return 0
  fi
  if [[ "${SKIP_KMD_BUILD:-0}" == "1" ]]; then
    # preloaded-KMD nodes (daemonset-preloaded.yaml): amdgpu is baked into
    # the image; only stage the userspace.
    [[ "${SKIP_PACKAGE_INSTALL:-0}" == "1" ]] || setup_repos
  else
    install_kernel_headers
    setup_repos
    build_and_load_kmd
  fi
  install_rocm_userspace
  verify_installation
  update_host_ld_cache
  update_cached_version
  echo "amdgpu + ROCm install complete"
}

# Source-guard: tests source this file and drive the functions directly
# against a scratch ROOT_MOUNT_DIR (tests/test_installer_exec.py).
if [[ "${BASH_SOURCE[0]:-}" == "$0" ]]; then
  main "$@"
fi
