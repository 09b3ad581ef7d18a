{{- define "wva.name" -}}
workload-variant-autoscaler
{{- end }}

{{- define "wva.fullname" -}}
{{ .Release.Name }}-workload-variant-autoscaler
{{- end }}
