/* Control library for the amd-smi preload-crash triage (ROUND2_NOTES
 * item 5): a preloadable .so that interposes NOTHING.  If the PLT-linked
 * amdsmi consumer crashes with THIS preloaded, the failure is a
 * loader-level effect of preloading per se (static-TLS pressure, link-map
 * ordering), not libvgpu-hip's hooks. */
int vgpu_empty_preload_marker = 1;
