/* Leveled stderr logging, env-controlled (LIBVGPU_LOG_LEVEL=0..3).
 * Reference analog: the hook's [4pdvGPU ...] leveled logging. */
#define _GNU_SOURCE
#include "vgpu.h"

#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <unistd.h>

static int g_level = -2; /* -2 = unread */

static const char *level_name[] = {"ERROR", "WARN", "INFO", "DEBUG"};

void vgpu_log(int level, const char *fmt, ...) {
  if (g_level == -2) {
    const char *e = getenv(ENV_LOG_LEVEL);
    g_level = e ? atoi(e) : VGPU_WARN;
  }
  if (level > g_level) return;
  char buf[1024];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  fprintf(stderr, "[vGPU-MI355X %s(%d)] %s\n", level_name[level & 3],
          (int)getpid(), buf);
}
