{{- define "amd-gpu.dp-image" -}}
{{ .Values.dp.image.repository }}:{{ .Values.dp.image.tag | default .Chart.AppVersion }}
{{- end -}}

{{- define "amd-gpu.labeller-image" -}}
{{ .Values.labeller.image.repository }}:{{ .Values.labeller.image.tag | default .Chart.AppVersion }}
{{- end -}}
