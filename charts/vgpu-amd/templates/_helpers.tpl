{{/* Name helpers */}}
{{- define "vgpu-amd.name" -}}
{{- default .Chart.Name .Values.nameOverride | trunc 63 | trimSuffix "-" -}}
{{- end -}}

{{- define "vgpu-amd.fullname" -}}
{{- if .Values.fullnameOverride -}}
{{- .Values.fullnameOverride | trunc 63 | trimSuffix "-" -}}
{{- else -}}
{{- printf "%s-%s" .Release.Name (include "vgpu-amd.name" .) | trunc 63 | trimSuffix "-" -}}
{{- end -}}
{{- end -}}

{{- define "vgpu-amd.scheduler" -}}
{{- printf "%s-scheduler" (include "vgpu-amd.fullname" .) | trunc 63 | trimSuffix "-" -}}
{{- end -}}

{{- define "vgpu-amd.device-plugin" -}}
{{- printf "%s-device-plugin" (include "vgpu-amd.fullname" .) | trunc 63 | trimSuffix "-" -}}
{{- end -}}

{{- define "vgpu-amd.chart" -}}
{{- printf "%s-%s" .Chart.Name .Chart.Version | replace "+" "_" | trunc 63 | trimSuffix "-" -}}
{{- end -}}

{{- define "vgpu-amd.labels" -}}
helm.sh/chart: {{ include "vgpu-amd.chart" . }}
{{ include "vgpu-amd.selectorLabels" . }}
app.kubernetes.io/version: {{ .Chart.AppVersion | quote }}
app.kubernetes.io/managed-by: {{ .Release.Service }}
{{- end -}}

{{- define "vgpu-amd.selectorLabels" -}}
app.kubernetes.io/name: {{ include "vgpu-amd.name" . }}
app.kubernetes.io/instance: {{ .Release.Name }}
{{- end -}}

{{- define "vgpu-amd.imagePullSecrets" -}}
{{- if .Values.imagePullSecrets }}
imagePullSecrets:
{{- range .Values.imagePullSecrets }}
  - name: {{ . }}
{{- end }}
{{- end }}
{{- end -}}
